import os, sys, torch
import torch.multiprocessing as mp

def worker(rank, world, port):
    import torch.distributed as dist
    def log(*a): print(f"[r{rank}]", *a, flush=True)
    os.environ["MASTER_ADDR"]="127.0.0.1"; os.environ["MASTER_PORT"]=str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    log("pg init")
    from pipegcn_amd.graph.halo import build_runtime_partition
    from pipegcn_amd.parallel.buffer import Buffer
    from pipegcn_amd.utils.timer import comm_timer
    from pipegcn_amd.graph import partition, synthetic
    tmp = "/tmp/dbg2r"
    u,v,n,ndata = synthetic.synth_global("small", nparts_hint=2, seed=5)
    if rank==0: partition.partition_and_save(tmp:=tmp,u=u,v=v,num_nodes=n,ndata=ndata,graph_dir=tmp,nparts=2) if False else partition.partition_and_save(u,v,n,ndata,tmp,2,"metis","vol",0)
    dist.barrier(); log("partitioned")
    part = partition.load_partition(tmp, rank)
    rp = build_runtime_partition(part, device="cuda:0")
    log("rp built")
    F=8
    buf = Buffer()
    buf.init_buffer(rp.num_in, rp.num_all, rp.boundary, rp.recv_shape,
                    [F, F], pipeline=True, backend="gloo", device="cuda:0",
                    corr_feat=True, corr_momentum=0.5)
    log("buffer init")
    peer = 1-rank; nhalo = rp.recv_shape[peer]
    avg = 0.0
    for epoch in range(4):
        feat = torch.full((rp.num_in, F), float(epoch+1+10*rank), device="cuda:0", requires_grad=(epoch>0))
        log("epoch", epoch, "update...")
        h = buf.update(1, feat)
        log("epoch", epoch, "update done")
        halo = h[rp.num_in:rp.num_in+nhalo]
        got = halo.detach().cpu()
        assert torch.allclose(got, torch.full((nhalo,F), avg), atol=1e-5), (epoch, got[0,0].item(), avg)
        if epoch > 0:
            h.sum().backward()
            log("epoch", epoch, "backward done")
        avg = 0.5*avg + 0.5*float(epoch+1+10*peer)
        buf.next_epoch(); comm_timer.clear()
    buf.synchronize(); buf.shutdown()
    log("DONE")

if __name__ == "__main__":
    mp.set_start_method("spawn")
    ps = [mp.Process(target=worker, args=(r, 2, 29771)) for r in range(2)]
    [p.start() for p in ps]
    import time
    t0=time.time()
    while time.time()-t0 < 300 and any(p.is_alive() for p in ps):
        time.sleep(2)
    for p in ps:
        if p.is_alive():
            print("TIMEOUT: terminating", p.pid, flush=True); p.terminate()
    print("exitcodes", [p.exitcode for p in ps], flush=True)
