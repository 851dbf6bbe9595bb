"""CLI smoke tests for the reference-compatible harnesses: one epoch of
AdHoc_train + one AdHoc_test pass on a committed 20-node case, checking the
exact reference CSV schemas and the checkpoint layout."""
import glob
import os

import pandas as pd
import pytest

CASE = "data_samples/aco_data_ba_demo/aco_case_seed500_m2_n20_s4.mat"


@pytest.mark.skipif(not os.path.isfile(CASE), reason="demo case not committed")
def test_adhoc_train_then_test_cli(tmp_path):
    from multihop_offload_amd.harness import adhoc_train, adhoc_test

    model_root = str(tmp_path / "model")
    out = str(tmp_path / "out")
    common_args = [
        "--datapath", os.path.dirname(CASE), "--limit_cases", "1",
        "--out", out, "--model_root", model_root,
        "--training_set", "SMOKE", "--instances", "2",
        "--seed", "11", "--device", "cpu",
    ]
    adhoc_train.main(common_args + ["--epochs", "1", "--batch", "2"])

    csvs = glob.glob(os.path.join(out, "aco_training_data_*.csv"))
    assert len(csvs) == 1
    df = pd.read_csv(csvs[0])
    assert list(df.columns) == adhoc_train.TRAIN_COLUMNS
    # methods cycle [baseline, local, GNN, GNN-test] per instance
    assert set(df["method"]) == {"baseline", "local", "GNN", "GNN-test"}
    assert len(df) == 2 * 4
    assert (df["tau"] > 0).all()

    # checkpoint written in the reference layout
    ckpts = glob.glob(os.path.join(
        model_root, "model_ChebConv_SMOKE*", "cp-*.ckpt*"))
    assert ckpts, os.listdir(model_root)

    adhoc_test.main(common_args)
    tcsvs = glob.glob(os.path.join(out, "Adhoc_test_*.csv"))
    assert len(tcsvs) == 1
    tdf = pd.read_csv(tcsvs[0])
    assert "Algo" in tdf.columns and set(tdf["Algo"]) >= {"baseline", "local"}


def test_demo_main(tmp_path):
    """The reference demo path (offloading_v3.main, stale/crashing as
    shipped) runs end-to-end here, including the metrics trace figure."""
    from multihop_offload_amd.harness import demo
    dec, est, emp = demo.main([
        "1", "--fig_dir", str(tmp_path / "fig"),
        "--out", str(tmp_path / "out"), "--trace"])
    assert len(dec) == len(est) == len(emp) == 5
    assert (emp > 0).all()
    figs = os.listdir(tmp_path / "fig")
    assert any("flow_routes" in f for f in figs)
    assert any("flow_packets" in f for f in figs)
    assert os.listdir(tmp_path / "out")
