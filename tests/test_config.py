"""Reference cfg-format compatibility (InputInfo, GraphSegment.cpp:222-292)."""
import textwrap

from neutronstarlite_amd.config import read_cfg


def test_reads_reference_shaped_cfg(tmp_path):
    # shape of /root/reference/gcn_reddit.cfg (content restated, not copied)
    cfg = tmp_path / "g.cfg"
    cfg.write_text(textwrap.dedent("""\
        ALGORITHM:GCN
        VERTICES:232965
        LAYERS:602-128-41
        EPOCHS:200
        EDGE_FILE:./data/reddit/reddit.edge.txt.bin
        FEATURE_FILE:./data/reddit/reddit.featuretablenorm_nts
        LABEL_FILE:./data/reddit/reddit.labeltablesorted
        MASK_FILE:./data/reddit/reddit.masksorted
        PROC_OVERLAP:1
        PROC_LOCAL:0
        PROC_CUDA:1
        PROC_REP:0
        LOCK_FREE:1
        LEARN_RATE:0.01
        WEIGHT_DECAY:0.0001
        DECAY_RATE:0.97
        DECAY_EPOCH:100
        DROP_RATE:0.5
    """))
    info = read_cfg(str(cfg))
    assert info.algorithm == "GCN"
    assert info.vertices == 232965
    assert info.layer_sizes == [602, 128, 41]
    assert info.edge_file.endswith("reddit.edge.txt.bin")
    assert info.process_overlap and info.with_cuda and info.lock_free
    assert not info.process_local
    assert info.learn_rate == 0.01 and info.decay_epoch == 100
    assert info.raw["DROP_RATE"] == "0.5"


def test_tolerates_blank_and_unknown_keys(tmp_path):
    cfg = tmp_path / "g.cfg"
    cfg.write_text("ALGORITHM:GAT\n\nSOME_FUTURE_KEY:7\nVERTICES:10\n")
    info = read_cfg(str(cfg))
    assert info.algorithm == "GAT" and info.vertices == 10
    assert info.raw["SOME_FUTURE_KEY"] == "7"
