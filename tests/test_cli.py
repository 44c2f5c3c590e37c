"""CLI surface parity with the reference (/root/reference/helper/parser.py):
every flag, both dash and underscore spellings, same defaults."""
from pipegcn_amd.cli import check_args, create_parser


def parse(argv):
    return create_parser(argv)


def test_reference_defaults():
    a = parse([])
    assert a.dataset == "reddit"
    assert a.model == "graphsage"
    assert a.dropout == 0.5
    assert a.lr == 1e-2
    assert a.n_epochs == 200
    assert a.n_partitions == 2
    assert a.n_hidden == 16
    assert a.n_layers == 2
    assert a.n_linear == 0
    assert a.norm == "layer"
    assert a.weight_decay == 0
    assert a.partition_obj == "vol"
    assert a.partition_method == "metis"
    assert a.enable_pipeline is False
    assert a.feat_corr is False and a.grad_corr is False
    assert a.corr_momentum == 0.95
    assert a.use_pp is False and a.inductive is False
    assert a.fix_seed is False and a.seed == 0
    assert a.log_every == 10
    assert a.backend == "gloo"
    assert a.port == 18118
    assert a.master_addr == "127.0.0.1"
    assert a.node_rank == 0 and a.parts_per_node == 10
    assert a.eval is True


def test_dual_spellings():
    a = parse(["--n_epochs", "7", "--n_hidden", "9", "--graph_name", "g",
               "--weight_decay", "0.1", "--partition_obj", "cut",
               "--partition_method", "random", "--enable_pipeline",
               "--feat_corr", "--grad_corr", "--corr_momentum", "0.5",
               "--use_pp", "--fix_seed", "--log_every", "3",
               "--master_addr", "10.0.0.1", "--node_rank", "1",
               "--parts_per_node", "4", "--n_feat", "5", "--n_class", "2",
               "--n_train", "11"])
    assert a.n_epochs == 7 and a.n_hidden == 9 and a.graph_name == "g"
    assert a.weight_decay == 0.1 and a.partition_obj == "cut"
    assert a.partition_method == "random" and a.enable_pipeline
    assert a.feat_corr and a.grad_corr and a.corr_momentum == 0.5
    assert a.use_pp and a.fix_seed and a.log_every == 3
    assert a.master_addr == "10.0.0.1" and a.node_rank == 1
    assert a.parts_per_node == 4
    assert (a.n_feat, a.n_class, a.n_train) == (5, 2, 11)


def test_norm_none_and_no_eval():
    a = check_args(parse(["--norm", "none", "--no-eval"]))
    assert a.norm is None
    assert a.eval is False
