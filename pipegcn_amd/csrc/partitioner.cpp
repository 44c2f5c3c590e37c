// Multilevel k-way graph partitioner (METIS-style; replaces the METIS
// library the reference used through dgl.distributed.partition_graph,
// /root/reference/helper/utils.py:143 — METIS is not available in this
// environment, so this is our own implementation):
//
//   1. COARSEN: heavy-edge matching collapses node pairs level by level
//      (edge weights accumulate; node weights = contained fine nodes).
//   2. INITIAL PARTITION on the coarsest graph: farthest-point seeded,
//      weight-balanced multi-source BFS growth.
//   3. UNCOARSEN: project the assignment back up, boundary-refining at
//      every level (FM-style single-node moves; the 'vol' objective uses
//      EXACT O(deg) communication-volume gains — the moving node's own
//      replica delta plus every neighbor's replica add/remove, tracked
//      with per-node neighbor-partition counters — whenever the counter
//      table fits PIPEGCN_PART_EXACT_VOL_MB (default 4096 MB; 111M nodes
//      x 8 parts = 3.4 GB), the reference's objtype='vol'; above the
//      limit it falls back to the approximate own-delta-only gain).
//
// Scale notes (papers100M: 111M nodes, multi-billion edges):
//   - level 0 dedups + strips self-loops into int32 arrays ONCE (the
//     symmetrized input duplicates every edge; duplicates would also
//     break the distinct-neighbor volume logic) and carries unit edge/
//     node weights implicitly (no 2x int32/int64 arrays at full scale);
//   - coarsening is a deterministic parallel two-pass; matching on big
//     levels is parallel Suitor with hash-de-tied keys; refinement is a
//     parallel-prefilter boundary queue with serial live application
//     (all thread-count-deterministic; measured in profiles/).
//
// PIPEGCN_PART_CHECK_VOL=1: verify every accepted vol move against a
// brute-force local volume recomputation (the exact-gain correctness
// oracle; used by tests, debug only).
//
// Exported as partition_graph_cpu with the original signature.

#include "common.h"

#include <ATen/Parallel.h>

#include <atomic>

#include <algorithm>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <deque>
#include <numeric>
#include <random>
#include <vector>

namespace {
struct PhaseTimer {
  bool on = std::getenv("PIPEGCN_PART_VERBOSE") != nullptr;
  std::chrono::steady_clock::time_point t =
      std::chrono::steady_clock::now();
  void lap(const char* name, int64_t n = -1) {
    if (!on) return;
    auto now = std::chrono::steady_clock::now();
    double ms = std::chrono::duration<double, std::milli>(now - t).count();
    fprintf(stderr, "[partition] %-18s %8.0f ms (n=%lld)\n", name, ms,
            (long long)n);
    t = now;
  }
};

int64_t env_int(const char* name, int64_t dflt) {
  const char* s = std::getenv(name);
  return s ? std::atoll(s) : dflt;
}
}  // namespace

namespace {

// CSR graph. Level 0 has implicit unit edge/node weights (ewp/nwp null)
// to stay lean at 100M+ nodes; coarse levels own their weight arrays.
struct Graph {
  int64_t n = 0;
  std::vector<int64_t> indptr;
  std::vector<int32_t> indices;
  std::vector<int32_t> ew_s;  // edge weights (empty => unit)
  std::vector<int64_t> nw_s;  // node weights (empty => unit)
  int64_t ew(int64_t e) const { return ew_s.empty() ? 1 : ew_s[e]; }
  int64_t nw(int64_t x) const { return nw_s.empty() ? 1 : nw_s[x]; }
};

// Parallel SUITOR matching (Manne-Bisseling) with hash-de-tied weights.
// Round-1 finding: on unit-weight graphs every proposal ties, each tie
// displaces a prior suitor, and every displaced node rescans its full
// neighbor list — a 20x slowdown at level 0. The fix (designed then,
// implemented now): a strict TOTAL order on edge keys,
//    key = (min(ew,0xFFFF)<<16 | hash16(u,v)) << 32 | proposer,
// so first proposals almost always stick and the final matching is the
// unique greedy matching of the de-tied weights — deterministic across
// thread counts and schedules by construction. Leftover unmatched nodes
// fall through to the same 2-hop pass as the serial path.
int64_t suitor_matching(const Graph& g, std::vector<int32_t>& match) {
  const int64_t n = g.n;
  std::vector<std::atomic<uint64_t>> suitor(n);
  at::parallel_for(0, n, 1 << 16, [&](int64_t b, int64_t en) {
    for (int64_t i = b; i < en; ++i)
      suitor[i].store(0, std::memory_order_relaxed);
  });
  auto edge_key = [&](int64_t u, int64_t v, int64_t ew) -> uint64_t {
    const uint64_t a = u < v ? u : v;
    const uint64_t bb = u < v ? v : u;
    uint64_t h = (a + 0x9E3779B97F4A7C15ull) * 0xBF58476D1CE4E5B9ull;
    h ^= (bb + 0x94D049BB133111EBull) * 0x2545F4914F6CDD1Dull;
    h ^= h >> 29;
    const uint64_t w = (uint64_t)std::min<int64_t>(ew, 0xFFFF);
    const uint64_t wkey = (w << 16) | (h & 0xFFFF);
    // +1 on the proposer so key 0 means "no suitor yet"
    return (wkey << 32) | (uint64_t)(u + 1);
  };
  at::parallel_for(0, n, 4096, [&](int64_t b, int64_t en) {
    for (int64_t start = b; start < en; ++start) {
      int64_t x = start;
      while (x >= 0) {
        // best eligible neighbor of x: highest key that still beats
        // v's current suitor
        int64_t best_v = -1;
        uint64_t best_key = 0;
        for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
          const int32_t v = g.indices[e];
          if (v == x) continue;
          const uint64_t k = edge_key(x, v, g.ew(e));
          if (k <= best_key) continue;
          if (k > suitor[v].load(std::memory_order_relaxed)) {
            best_key = k;
            best_v = v;
          }
        }
        if (best_v < 0) break;
        uint64_t old = suitor[best_v].load(std::memory_order_relaxed);
        if (old >= best_key) continue;  // raced: re-scan x
        if (suitor[best_v].compare_exchange_weak(
                old, best_key, std::memory_order_acq_rel)) {
          // displaced proposer (if any) must re-propose elsewhere
          x = (int64_t)(old & 0xFFFFFFFFull) - 1;
        }
        // CAS failure: loop re-scans x against the fresher state
      }
    }
  });
  // mutual suitors are matched; everyone else stays -1 for the 2-hop pass
  match.assign(n, -1);
  at::parallel_for(0, n, 1 << 16, [&](int64_t b, int64_t en) {
    for (int64_t u = b; u < en; ++u) {
      const uint64_t su = suitor[u].load(std::memory_order_relaxed);
      const int64_t p = (int64_t)(su & 0xFFFFFFFFull) - 1;
      if (p < 0) continue;
      const uint64_t sp = suitor[p].load(std::memory_order_relaxed);
      if ((int64_t)(sp & 0xFFFFFFFFull) - 1 == u)
        match[u] = (int32_t)p;  // mutual (written from both sides)
    }
  });
  int64_t matched = 0;
  for (int64_t u = 0; u < n; ++u)
    if (match[u] >= 0) matched++;
  return matched;
}

// heavy-edge matching; returns coarse count and fine->coarse map
int64_t heavy_edge_matching(const Graph& g, std::mt19937_64& rng,
                            std::vector<int32_t>& cmap) {
  const int64_t n = g.n;
  std::vector<int32_t> match;
  const int64_t suitor_min = env_int("PIPEGCN_PART_SUITOR_MIN", 2000000);
  if (n >= suitor_min) {
    // big levels: parallel Suitor with hash-de-tied keys (see above)
    suitor_matching(g, match);
  } else {
    match.assign(n, -1);
    std::vector<int32_t> order(n);
    std::iota(order.begin(), order.end(), 0);
    std::shuffle(order.begin(), order.end(), rng);
    for (int32_t u : order) {
      if (match[u] >= 0) continue;
      int32_t best = -1;
      int64_t best_w = -1;
      for (int64_t e = g.indptr[u]; e < g.indptr[u + 1]; ++e) {
        const int32_t v = g.indices[e];
        if (v == u || match[v] >= 0) continue;
        if (g.ew(e) > best_w) {
          best_w = g.ew(e);
          best = v;
        }
      }
      if (best >= 0) {
        match[u] = best;
        match[best] = u;
      }
    }
  }
  // 2-hop pass (METIS-style): heavy-tailed graphs stall the 1-hop pass —
  // once the hubs are matched, their leaves have no unmatched neighbor
  // left. Merge pairs of unmatched leaves that SHARE a neighbor: one
  // deterministic O(E) sweep with a per-node pending-leaf slot.
  {
    std::vector<int32_t> pending(n, -1);
    for (int64_t u = 0; u < n; ++u) {
      if (match[u] >= 0) continue;
      bool done = false;
      for (int64_t e = g.indptr[u]; e < g.indptr[u + 1] && !done; ++e) {
        const int32_t v = g.indices[e];
        if (v == u) continue;
        const int32_t w = pending[v];
        if (w >= 0 && w != (int32_t)u && match[w] < 0) {
          match[u] = w;
          match[w] = (int32_t)u;
          pending[v] = -1;
          done = true;
        } else {
          pending[v] = (int32_t)u;
        }
      }
    }
    for (int64_t u = 0; u < n; ++u)
      if (match[u] < 0) match[u] = (int32_t)u;
  }
  cmap.assign(n, -1);
  int64_t nc = 0;
  for (int64_t u = 0; u < n; ++u) {
    if (cmap[u] >= 0) continue;
    cmap[u] = static_cast<int32_t>(nc);
    const int32_t v = match[u];
    if (v >= 0 && v != u && cmap[v] < 0) cmap[v] = static_cast<int32_t>(nc);
    nc++;
  }
  return nc;
}

Graph coarsen(const Graph& g, const std::vector<int32_t>& cmap, int64_t nc) {
  // Deterministic parallel two-pass construction (the dominant cost at
  // large scale): pass 1 counts each coarse node's distinct neighbors,
  // pass 2 fills preallocated arrays — per-chunk scratch mark arrays, no
  // locks, output independent of the thread count.
  Graph c;
  c.n = nc;
  c.nw_s.assign(nc, 0);
  // members of each coarse node (at most 2)
  std::vector<int32_t> m1(nc, -1), m2(nc, -1);
  for (int64_t u = 0; u < g.n; ++u) {
    const int32_t cu = cmap[u];
    c.nw_s[cu] += g.nw(u);
    if (m1[cu] < 0)
      m1[cu] = static_cast<int32_t>(u);
    else
      m2[cu] = static_cast<int32_t>(u);
  }
  c.indptr.assign(nc + 1, 0);
  const int64_t grain = std::max<int64_t>(4096, nc / 64);
  at::parallel_for(0, nc, grain, [&](int64_t b, int64_t en) {
    std::vector<int32_t> mark(nc, -1);
    for (int64_t cu = b; cu < en; ++cu) {
      int64_t deg = 0;
      for (int32_t u : {m1[cu], m2[cu]}) {
        if (u < 0) continue;
        for (int64_t e = g.indptr[u]; e < g.indptr[u + 1]; ++e) {
          const int32_t cv = cmap[g.indices[e]];
          if (cv == cu) continue;
          if (mark[cv] != static_cast<int32_t>(cu)) {
            mark[cv] = static_cast<int32_t>(cu);
            deg++;
          }
        }
      }
      c.indptr[cu + 1] = deg;
    }
  });
  for (int64_t i = 0; i < nc; ++i) c.indptr[i + 1] += c.indptr[i];
  c.indices.resize(c.indptr[nc]);
  c.ew_s.resize(c.indptr[nc]);
  at::parallel_for(0, nc, grain, [&](int64_t b, int64_t en) {
    std::vector<int32_t> mark(nc, -1);
    std::vector<int64_t> slot(nc, 0);
    for (int64_t cu = b; cu < en; ++cu) {
      int64_t w = c.indptr[cu];
      for (int32_t u : {m1[cu], m2[cu]}) {
        if (u < 0) continue;
        for (int64_t e = g.indptr[u]; e < g.indptr[u + 1]; ++e) {
          const int32_t cv = cmap[g.indices[e]];
          if (cv == cu) continue;
          if (mark[cv] != static_cast<int32_t>(cu)) {
            mark[cv] = static_cast<int32_t>(cu);
            slot[cv] = w;
            c.indices[w] = cv;
            c.ew_s[w] = static_cast<int32_t>(g.ew(e));
            w++;
          } else {
            c.ew_s[slot[cv]] += static_cast<int32_t>(g.ew(e));
          }
        }
      }
    }
  });
  return c;
}

// weight-balanced initial partition: farthest-point seeds + BFS growth
void initial_partition(const Graph& g, int64_t nparts, int64_t cap_w,
                       std::mt19937_64& rng, std::vector<int32_t>& part) {
  const int64_t n = g.n;
  part.assign(n, -1);
  std::vector<int64_t> seeds;
  {
    std::uniform_int_distribution<int64_t> uni(0, n - 1);
    seeds.push_back(uni(rng));
    std::vector<int32_t> dist(n);
    for (int64_t k = 1; k < nparts; ++k) {
      std::fill(dist.begin(), dist.end(), -1);
      std::deque<int64_t> q;
      for (int64_t s : seeds) {
        dist[s] = 0;
        q.push_back(s);
      }
      int64_t far = seeds[0];
      while (!q.empty()) {
        const int64_t x = q.front();
        q.pop_front();
        far = x;
        for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
          const int32_t y = g.indices[e];
          if (dist[y] < 0) {
            dist[y] = dist[x] + 1;
            q.push_back(y);
          }
        }
      }
      int64_t pick = -1;
      for (int64_t x = 0; x < n; ++x)
        if (dist[x] < 0) {
          pick = x;
          break;
        }
      seeds.push_back(pick < 0 ? far : pick);
    }
  }
  std::vector<int64_t> psize(nparts, 0);
  std::vector<std::deque<int64_t>> frontier(nparts);
  for (int64_t k = 0; k < nparts; ++k) {
    const int64_t s = seeds[k];
    if (part[s] < 0) {
      part[s] = static_cast<int32_t>(k);
      psize[k] += g.nw(s);
      frontier[k].push_back(s);
    }
  }
  bool progress = true;
  while (progress) {
    progress = false;
    for (int64_t k = 0; k < nparts; ++k) {
      if (psize[k] >= cap_w) continue;
      while (!frontier[k].empty() && psize[k] < cap_w) {
        const int64_t x = frontier[k].front();
        bool claimed = false;
        for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
          const int32_t y = g.indices[e];
          if (part[y] < 0) {
            part[y] = static_cast<int32_t>(k);
            psize[k] += g.nw(y);
            frontier[k].push_back(y);
            claimed = true;
            progress = true;
            break;
          }
        }
        if (!claimed)
          frontier[k].pop_front();
        else
          break;  // one claim per turn keeps growth balanced
      }
    }
  }
  for (int64_t x = 0; x < n; ++x) {
    if (part[x] < 0) {
      const int64_t k =
          std::min_element(psize.begin(), psize.end()) - psize.begin();
      part[x] = static_cast<int32_t>(k);
      psize[k] += g.nw(x);
    }
  }
}

// brute-force volume term of one node: nw(v) * #{p != part[v] : some
// neighbor of v is in p}  (the CHECK_VOL oracle)
int64_t node_vol(const Graph& g, const std::vector<int32_t>& part,
                 std::vector<int64_t>& scratch, int64_t v) {
  int64_t reps = 0;
  for (int64_t e = g.indptr[v]; e < g.indptr[v + 1]; ++e) {
    const int32_t u = g.indices[e];
    if (u == v) continue;
    const int32_t p = part[u];
    if (p != part[v] && scratch[p] == 0) {
      scratch[p] = 1;
      reps++;
    }
  }
  for (int64_t e = g.indptr[v]; e < g.indptr[v + 1]; ++e)
    scratch[part[g.indices[e]]] = 0;
  return g.nw(v) * reps;
}

// FM-style boundary refinement (single-node moves), parallel-evaluated:
// rounds of read-only gain evaluation over the active-queue snapshot
// (at::parallel_for against the frozen part[]), then serial in-order
// application that re-queues any candidate whose neighborhood changed
// earlier in the same round (staleness guard). Deterministic for a fixed
// graph+seed regardless of thread count: evaluation is a pure function
// of the round snapshot and application order is node order.
//  objective 0: weighted edge-cut gains.
//  objective 1 ("vol"): exact communication-volume gains when the
//    nbrcnt table fits (see header comment); approximate otherwise.
void refine(const Graph& g, int64_t nparts, int64_t objective, int64_t lo_w,
            int64_t cap_w, int64_t passes, std::vector<int32_t>& part) {
  const int64_t n = g.n;
  std::vector<int64_t> psize(nparts, 0);
  for (int64_t x = 0; x < n; ++x) psize[part[x]] += g.nw(x);

  // exact-vol neighbor-partition counters: nbrcnt[v*nparts+p] = number of
  // v's incident edges leading into partition p (level-0 graphs are
  // deduped+loop-free, coarse graphs are simple => edge count == distinct
  // neighbor count, which is what volume semantics need)
  bool exact = false;
  std::vector<int32_t> nbrcnt;
  if (objective == 1) {
    const int64_t limit_b =
        env_int("PIPEGCN_PART_EXACT_VOL_MB", 4096) * 1024 * 1024;
    if (n * nparts * (int64_t)sizeof(int32_t) <= limit_b) {
      exact = true;
      nbrcnt.assign(n * nparts, 0);
      at::parallel_for(0, n, 8192, [&](int64_t b, int64_t en) {
        for (int64_t x = b; x < en; ++x)
          for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
            const int32_t v = g.indices[e];
            if (v != x) nbrcnt[x * nparts + part[v]]++;
          }
      });
    }
  }
  const bool check =
      exact && std::getenv("PIPEGCN_PART_CHECK_VOL") != nullptr;
  std::vector<int64_t> vol_scratch;
  if (check) vol_scratch.assign(nparts, 0);

  struct Scratch {
    std::vector<int64_t> cnt, addq;
    std::vector<int32_t> touched;
  };

  // gain evaluation for one node against the current (frozen) part[]:
  // fills best target (== part[x] when no improving move) and its gains
  auto evaluate = [&](int64_t x, Scratch& s, int32_t& out_best,
                      int64_t& out_vol, int64_t& out_cut) {
    const int32_t a = part[x];
    out_best = a;
    out_vol = out_cut = 0;
    if (psize[a] - g.nw(x) < lo_w) return;
    bool boundary = false;
    for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
      const int32_t v = g.indices[e];
      if (v == x) continue;  // self-loops carry no cut/volume
      const int32_t q = part[v];
      if (s.cnt[q] == 0) s.touched.push_back(q);
      s.cnt[q] += g.ew(e);
      if (q != a) boundary = true;
    }
    if (boundary) {
      int64_t rep_a = 0;  // replicas of x if it stays in a
      if (objective == 1)
        for (const int32_t t : s.touched)
          if (t != a && s.cnt[t] > 0) rep_a++;
      int64_t rem_a = 0;  // nbr volume freed when x leaves a
      if (exact) {
        for (const int32_t t : s.touched) s.addq[t] = 0;
        for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
          const int32_t v = g.indices[e];
          if (v == x) continue;
          const int32_t b = part[v];
          const int32_t* row = nbrcnt.data() + (int64_t)v * nparts;
          if (b != a && row[a] == 1) rem_a += g.nw(v);
          for (const int32_t t : s.touched)
            if (t != b && row[t] == 0) s.addq[t] += g.nw(v);
        }
      }
      int64_t best_gain = 0;
      for (const int32_t q : s.touched) {
        if (q == a || psize[q] + g.nw(x) > cap_w) continue;
        const int64_t cut_gain = s.cnt[q] - s.cnt[a];
        if (objective == 1) {
          int64_t rep_q = 0;
          for (const int32_t t : s.touched)
            if (t != q && s.cnt[t] > 0) rep_q++;
          if (exact) {
            // exact volume gain: own replicas + neighbor add/remove
            const int64_t vol_gain =
                g.nw(x) * (rep_a - rep_q) + rem_a - s.addq[q];
            // lexicographic (volume, cut): volume is the objective,
            // cut breaks ties; only strictly-improving moves
            // (vol>0, or vol-neutral with cut>0) are ever accepted
            if ((vol_gain > out_vol ||
                 (vol_gain == out_vol && cut_gain > out_cut)) &&
                (vol_gain > 0 || (vol_gain == 0 && cut_gain > 0))) {
              out_best = q;
              out_vol = vol_gain;
              out_cut = cut_gain;
            }
            continue;
          }
          // approximate: own replica delta only
          const int64_t gain = cut_gain + rep_a - rep_q;
          if (gain > best_gain) {
            best_gain = gain;
            out_best = q;
          }
          continue;
        }
        if (cut_gain > best_gain) {
          best_gain = cut_gain;
          out_best = q;
        }
      }
    }
    for (const int32_t q : s.touched) s.cnt[q] = 0;
    s.touched.clear();
  };

  // active queue, seeded with every boundary node in node order
  // (parallel detection, chunk-ordered merge keeps it deterministic)
  std::vector<uint8_t> in_q(n, 0);
  std::vector<int64_t> work, next;
  {
    const int64_t nchunk = 64;
    const int64_t csz = (n + nchunk - 1) / nchunk;
    std::vector<std::vector<int64_t>> found(nchunk);
    at::parallel_for(0, nchunk, 1, [&](int64_t cb, int64_t ce) {
      for (int64_t c = cb; c < ce; ++c) {
        for (int64_t x = c * csz; x < std::min(n, (c + 1) * csz); ++x)
          for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e)
            if (g.indices[e] != x && part[g.indices[e]] != part[x]) {
              found[c].push_back(x);
              break;
            }
      }
    });
    for (auto& f : found)
      for (int64_t x : f) {
        work.push_back(x);
        in_q[x] = 1;
      }
  }
  std::vector<int32_t> bestv;
  // evaluation budget guards the approximate mode (its gain is heuristic,
  // so the true objective need not strictly decrease); exact mode
  // terminates on its own (lexicographic (vol,cut) strictly decreases)
  int64_t pops_left = std::max<int64_t>(passes, 4) * std::max<int64_t>(n, 1);
  // `passes` doubles as the round cap: each round is one parallel sweep
  // of the active queue (the FM-pass analogue); refinement gains decay
  // fast with rounds, so the cap trades a few % of volume for bounded
  // O(rounds * boundary) work at 100M+ nodes
  const int64_t cap_env = env_int("PIPEGCN_PART_ROUNDS", 0);
  const int64_t rounds_cap =
      cap_env > 0 ? cap_env
                  : (passes >= 999 ? (int64_t)1 << 40
                                   : std::max<int64_t>(passes, 2));
  int64_t n_rounds = 0, n_pre = 0, n_live = 0, n_moves = 0;
  double t_par = 0, t_ser = 0;
  using clk = std::chrono::steady_clock;

  while (!work.empty() && pops_left > 0 && n_rounds < rounds_cap) {
    ++n_rounds;
    const int64_t R = std::min<int64_t>((int64_t)work.size(), pops_left);
    pops_left -= R;
    bestv.assign(R, 0);
    // parallel PREFILTER: gains against the round-start snapshot select
    // candidates; the serial pass below re-evaluates each candidate on
    // the LIVE state before applying, so a stale prefilter result can
    // cost a wasted re-evaluation but never a wrong move. Small rounds
    // run inline (parallel_for launch overhead dwarfs tiny batches).
    auto eval_range = [&](int64_t b, int64_t en) {
      Scratch s;
      s.cnt.assign(nparts, 0);
      s.addq.assign(nparts, 0);
      s.touched.reserve(64);
      int64_t vol_g, cut_g;
      for (int64_t i = b; i < en; ++i)
        evaluate(work[i], s, bestv[i], vol_g, cut_g);
    };
    auto tp0 = clk::now();
    if (R < 8192)
      eval_range(0, R);
    else
      at::parallel_for(0, R, 4096, eval_range);
    auto tp1 = clk::now();
    t_par += std::chrono::duration<double>(tp1 - tp0).count();
    next.clear();
    auto push_next = [&](int64_t v) {
      if (!in_q[v]) {
        in_q[v] = 1;
        next.push_back(v);
      }
    };
    Scratch s;
    s.cnt.assign(nparts, 0);
    s.addq.assign(nparts, 0);
    s.touched.reserve(64);
    for (int64_t i = 0; i < R; ++i) {
      const int64_t x = work[i];
      in_q[x] = 0;
      if (bestv[i] == part[x]) continue;  // prefilter: no move candidate
      // live re-evaluation (neighbors/psize may have changed since the
      // snapshot; exact gains also depend on 2-hop state via the
      // neighbors' counter rows, so only a live gain is trustworthy)
      const int32_t a = part[x];
      int32_t best;
      int64_t vol_gain, cut_gain;
      ++n_live;
      evaluate(x, s, best, vol_gain, cut_gain);
      if (best == a) continue;
      ++n_moves;
      if (check) {
        // oracle: recompute local volume of {x} u N(x) before/after
        int64_t before = node_vol(g, part, vol_scratch, x);
        for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e)
          if (g.indices[e] != x)
            before += node_vol(g, part, vol_scratch, g.indices[e]);
        part[x] = best;
        int64_t after = node_vol(g, part, vol_scratch, x);
        for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e)
          if (g.indices[e] != x)
            after += node_vol(g, part, vol_scratch, g.indices[e]);
        part[x] = a;
        TORCH_CHECK(before - after == vol_gain,
                    "exact vol gain mismatch: predicted ", vol_gain,
                    " actual ", before - after, " at node ", x);
      }
      if (exact) {
        // maintain neighbor counters for the move a -> best
        for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
          const int32_t v = g.indices[e];
          if (v == x) continue;
          nbrcnt[(int64_t)v * nparts + a]--;
          nbrcnt[(int64_t)v * nparts + best]++;
        }
      }
      part[x] = best;
      psize[a] -= g.nw(x);
      psize[best] += g.nw(x);
      // the move changed every neighbor's gain landscape (and x's own)
      for (int64_t e = g.indptr[x]; e < g.indptr[x + 1]; ++e) {
        const int32_t v = g.indices[e];
        if (v != x) push_next(v);
      }
      push_next(x);
    }
    n_pre += R;
    t_ser += std::chrono::duration<double>(clk::now() - tp1).count();
    std::swap(work, next);
  }
  if (std::getenv("PIPEGCN_PART_VERBOSE"))
    fprintf(stderr,
            "[partition]   refine n=%lld rounds=%lld pre=%lld live=%lld "
            "moves=%lld par=%.2fs ser=%.2fs\n",
            (long long)n, (long long)n_rounds, (long long)n_pre,
            (long long)n_live, (long long)n_moves, t_par, t_ser);
}

}  // namespace

torch::Tensor partition_graph_cpu(torch::Tensor indptr, torch::Tensor indices,
                                  int64_t nparts, int64_t objective,
                                  double balance_slack, int64_t n_refine_passes,
                                  int64_t seed) {
  TORCH_CHECK(nparts >= 1, "nparts must be >= 1");
  const int64_t N = indptr.numel() - 1;
  auto out = torch::zeros({N}, torch::kInt);
  if (nparts == 1 || N == 0) return out;

  std::mt19937_64 rng(seed);

  // level 0: dedup + strip self-loops ONCE (the symmetrized input carries
  // every undirected edge twice; duplicate edges would break the
  // distinct-neighbor volume counters), unit weights implicit.
  std::vector<Graph> levels(1);
  {
    Graph& g0 = levels[0];
    g0.n = N;
    const int64_t* ip = indptr.data_ptr<int64_t>();
    const int32_t* xp = indices.data_ptr<int32_t>();
    g0.indptr.assign(N + 1, 0);
    at::parallel_for(0, N, 8192, [&](int64_t b, int64_t en) {
      std::vector<int32_t> row;
      for (int64_t x = b; x < en; ++x) {
        row.assign(xp + ip[x], xp + ip[x + 1]);
        std::sort(row.begin(), row.end());
        int64_t deg = 0;
        int32_t prev = -1;
        for (const int32_t v : row)
          if (v != x && v != prev) {
            prev = v;
            deg++;
          }
        g0.indptr[x + 1] = deg;
      }
    });
    for (int64_t i = 0; i < N; ++i) g0.indptr[i + 1] += g0.indptr[i];
    g0.indices.resize(g0.indptr[N]);
    at::parallel_for(0, N, 8192, [&](int64_t b, int64_t en) {
      std::vector<int32_t> row;
      for (int64_t x = b; x < en; ++x) {
        row.assign(xp + ip[x], xp + ip[x + 1]);
        std::sort(row.begin(), row.end());
        int64_t w = g0.indptr[x];
        int32_t prev = -1;
        for (const int32_t v : row)
          if (v != x && v != prev) {
            prev = v;
            g0.indices[w++] = v;
          }
      }
    });
  }
  std::vector<std::vector<int32_t>> cmaps;
  PhaseTimer pt;
  pt.lap("level0 dedup", N);

  // --- coarsen (memory-bounded level retention)
  // stop coarsening early: community graphs densify as they coarsen
  // (E shrinks much slower than n), which makes both coarsen() and the
  // per-eval O(deg) refinement cost blow up on the small-n levels, and
  // every extra level adds projection error the finer levels must repair.
  // Dense MIDDLE levels of a 100M+-node graph can hold >1B edges EACH;
  // retaining every one for uncoarsening refinement OOMs long before the
  // coarsest solve (measured: 65 GB RSS at papers100M scale). Levels
  // whose edge count exceeds PIPEGCN_PART_KEEP_EDGES (default 400M) are
  // used for matching and then DISCARDED: their projection maps compose,
  // and refinement happens only at the retained levels (level 0 always).
  const int64_t coarse_target = std::max<int64_t>(
      env_int("PIPEGCN_PART_COARSE_N", 65536), 128 * nparts);
  const int64_t keep_edges =
      env_int("PIPEGCN_PART_KEEP_EDGES", 400000000);
  {
    std::vector<int32_t> pending;  // levels.back() ids -> front ids
    Graph front_store;             // the front when it is not retained
    int64_t n_levels_total = 1;
    while (true) {
      Graph& front = pending.empty() ? levels.back() : front_store;
      if (front.n <= coarse_target || n_levels_total >= 24) break;
      std::vector<int32_t> cmap;
      const int64_t nc = heavy_edge_matching(front, rng, cmap);
      if (nc > front.n * 95 / 100) break;  // matching stalled
      const int64_t e_prev = (int64_t)front.indices.size();
      Graph next = coarsen(front, cmap, nc);
      n_levels_total++;
      // densification guard: stop when halving n no longer sheds edges
      const bool densified =
          (int64_t)next.indices.size() > e_prev * 85 / 100 &&
          nc < 4 * coarse_target;
      if (pending.empty()) {
        pending = std::move(cmap);
      } else {
        at::parallel_for(0, (int64_t)pending.size(), 1 << 16,
                         [&](int64_t b, int64_t en) {
                           for (int64_t i = b; i < en; ++i)
                             pending[i] = cmap[pending[i]];
                         });
      }
      const bool retain = (int64_t)next.indices.size() <= keep_edges ||
                          densified;
      if (retain) {
        levels.push_back(std::move(next));
        cmaps.push_back(std::move(pending));
        pending.clear();
        front_store = Graph();  // free any unretained front
      } else {
        front_store = std::move(next);
      }
      pt.lap(retain ? "match+coarsen" : "match+coarsen drop", nc);
      if (densified) break;
    }
    // the coarsening front is the coarsest-solve target: retain it
    if (!pending.empty()) {
      levels.push_back(std::move(front_store));
      cmaps.push_back(std::move(pending));
    }
  }

  // --- initial partition on the coarsest level
  const int64_t tot_w = N;
  const int64_t cap_w =
      static_cast<int64_t>((double)tot_w / nparts * (1.0 + balance_slack)) + 1;
  const int64_t lo_w =
      static_cast<int64_t>((double)tot_w / nparts * (1.0 - balance_slack));
  // Random restarts at the coarsest level (it is tiny — a few thousand
  // nodes): greedy single-node refinement cannot escape a bad seeded
  // growth, so try several and keep the lowest weighted edge cut.
  std::vector<int32_t> part;
  {
    const Graph& gc = levels.back();
    auto cut_of = [&](const std::vector<int32_t>& p) {
      int64_t cut = 0;
      for (int64_t x = 0; x < gc.n; ++x)
        for (int64_t e = gc.indptr[x]; e < gc.indptr[x + 1]; ++e)
          if (p[gc.indices[e]] != p[x]) cut += gc.ew(e);
      return cut;
    };
    // restart count scales down with coarsest size (matching can stall
    // well above the target on heavy-tailed graphs; 8 full restarts on a
    // 100k+-node coarsest level would dominate the whole partition)
    const int64_t trials =
        std::max<int64_t>(2, std::min<int64_t>(8, 8 * 8192 / gc.n));
    int64_t best_cut = -1;
    for (int64_t trial = 0; trial < trials; ++trial) {
      std::vector<int32_t> cand;
      initial_partition(gc, nparts, cap_w, rng, cand);
      refine(gc, nparts, objective, lo_w, cap_w, 10, cand);
      const int64_t c = cut_of(cand);
      if (best_cut < 0 || c < best_cut) {
        best_cut = c;
        part = std::move(cand);
      }
    }
    // polish only the winner with the full budget
    refine(gc, nparts, objective, lo_w, cap_w,
           std::max<int64_t>(n_refine_passes, 32), part);
  }

  pt.lap("coarsest solve", levels.back().n);

  // --- uncoarsen + refine each level
  for (int64_t lvl = static_cast<int64_t>(levels.size()) - 2; lvl >= 0;
       --lvl) {
    const std::vector<int32_t>& cmap = cmaps[lvl];
    std::vector<int32_t> fine(levels[lvl].n);
    for (int64_t u = 0; u < levels[lvl].n; ++u) fine[u] = part[cmap[u]];
    part = std::move(fine);
    const int64_t passes =
        lvl == 0 ? std::max<int64_t>(n_refine_passes / 2, 4)
                 : std::max<int64_t>(n_refine_passes, 16);
    refine(levels[lvl], nparts, objective, lo_w, cap_w, passes, part);
    pt.lap("project+refine", levels[lvl].n);
    levels.pop_back();  // the finer level is no longer needed
  }

  std::memcpy(out.data_ptr<int32_t>(), part.data(), N * sizeof(int32_t));
  return out;
}
