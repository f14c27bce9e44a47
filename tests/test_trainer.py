"""Trainer CLI + harness tests: LR schedules, CSV schema + plotting
round-trip, ClusterManager checkpointing, end-to-end 2-rank CLI run."""

import os
import subprocess
import sys
import types

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def make_args(**kw):
    import gossip_sgd

    defaults = dict(
        lr=0.1, batch_size=32, world_size=8, warmup=True,
        lr_schedule={30: 0.1, 60: 0.1, 80: 0.1},
    )
    defaults.update(kw)
    return types.SimpleNamespace(**defaults)


class FakeOpt:
    def __init__(self):
        self.param_groups = [{"lr": 0.0}]


def test_lr_warmup_and_decay():
    from gossip_sgd import update_learning_rate

    args = make_args()
    opt = FakeOpt()
    target = 0.1 * 32 * 8 / 256  # = 0.1

    # warmup epoch 0, itr 0 (target <= ref lr -> jump straight to target)
    lr0 = update_learning_rate(args, opt, epoch=0, itr=0, itr_per_epoch=100)
    assert lr0 == pytest.approx(target)

    # larger world: ramp linearly over 5 epochs
    args2 = make_args(world_size=32)
    target2 = 0.1 * 32 * 32 / 256  # 0.4
    lr_start = update_learning_rate(args2, opt, 0, itr=0, itr_per_epoch=100)
    assert 0.1 < lr_start < target2
    lr_end = update_learning_rate(args2, opt, 4, itr=99, itr_per_epoch=100)
    assert lr_end == pytest.approx(target2, rel=1e-3)

    # after warmup, decay at 30/60/80
    assert update_learning_rate(args2, opt, 10) == pytest.approx(target2)
    assert update_learning_rate(args2, opt, 35) == pytest.approx(target2 * 0.1)
    assert update_learning_rate(args2, opt, 65) == pytest.approx(
        target2 * 0.01
    )
    assert update_learning_rate(args2, opt, 85) == pytest.approx(
        target2 * 0.001
    )


def test_pairs_to_dict():
    from gossip_sgd import pairs_to_dict

    assert pairs_to_dict([30, 0.1, 60, 0.2], None) == {30: 0.1, 60: 0.2}
    assert pairs_to_dict(None, [0, 1]) == {0: 1}


def test_bilat_lr_uses_global_epoch():
    from gossip_sgd_adpsgd import compute_bilat_lr

    args = make_args(global_epoch=35, global_itr=1000, world_size=8)
    lr = compute_bilat_lr(args, itr_per_epoch=100)
    assert lr == pytest.approx(0.1 * 32 * 8 / 256 * 0.1)


def test_cluster_manager_checkpoint(tmp_path):
    from stochastic_gradient_push_amd.utils.cluster_manager import (
        ClusterManager,
    )

    ClusterManager.set_checkpoint_dir(str(tmp_path) + "/")
    state = {"is_best": True, "x": torch.tensor([1.0])}
    cm = ClusterManager(rank=0, world_size=1, state=state, all_workers=True)
    cm.save_checkpoint(requeue_on_signal=False)
    assert os.path.exists(cm.checkpoint_fpath)
    assert os.path.exists(cm.model_best_fpath)  # is_best copied
    loaded = torch.load(cm.checkpoint_fpath, weights_only=False)
    assert torch.equal(loaded["x"], torch.tensor([1.0]))
    # epoch-tagged checkpoints
    state["is_best"] = False
    cm.save_checkpoint(epoch_id=3, requeue_on_signal=False)
    assert os.path.exists(
        str(tmp_path) + "/ep3_" + cm.checkpoint_fname
    )


def _run_trainer(tmp, extra, script="gossip_sgd.py", nprocs=1, timeout=240):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    base = [
        sys.executable, os.path.join(REPO, script),
        "--num_epochs", "1",
        "--num_iterations_per_training_epoch", "2",
        "--batch_size", "2", "--synthetic_size", "8",
        "--model", "resnet18", "--num_classes", "10",
        "--image_size", "32", "--num_dataloader_workers", "0",
        "--device", "cpu", "--checkpoint_dir", f"{tmp}/ckpt/",
        "--num_itr_ignore", "0", "--backend", "gloo",
        "--network_interface_type", "auto",
    ] + extra
    if nprocs == 1:
        env["RANK"] = "0"
        env["WORLD_SIZE"] = "1"
        subprocess.run(base, env=env, check=True, timeout=timeout,
                       cwd=str(tmp))
    else:
        procs = []
        try:
            for r in range(nprocs):
                e = dict(env)
                e["RANK"] = str(r)
                e["WORLD_SIZE"] = str(nprocs)
                procs.append(subprocess.Popen(base, env=e, cwd=str(tmp)))
            for p in procs:
                assert p.wait(timeout=timeout) == 0
        finally:
            # never leak rank processes on failure: a survivor keeps
            # the rendezvous port bound and cascades into later tests
            for p in procs:
                if p.poll() is None:
                    p.kill()
            for p in procs:
                try:
                    p.wait(timeout=10)
                except Exception:
                    pass


def test_trainer_cli_single_process(tmp_path):
    _run_trainer(tmp_path, ["--graph_type", "-1", "--master_port", "29811"])
    csv = f"{tmp_path}/ckpt/out_r0_n1.csv"
    assert os.path.exists(csv)

    from visualization.plotting import load_experiment, summarize

    runs = load_experiment(f"{tmp_path}/ckpt/")
    assert 0 in runs
    summary = summarize(runs)
    assert summary.iloc[0]["world_size"] == 1
    assert summary.iloc[0]["best_val_prec1"] is not None


def test_trainer_cli_two_rank_sgp(tmp_path):
    _run_trainer(
        tmp_path, ["--push_sum", "True", "--master_port", "29812"],
        nprocs=2,
    )
    assert os.path.exists(f"{tmp_path}/ckpt/out_r0_n2.csv")
    assert os.path.exists(f"{tmp_path}/ckpt/checkpoint_r1_n2.pth.tar")

    from visualization.plotting import load_experiment

    runs = load_experiment(f"{tmp_path}/ckpt/")
    assert set(runs) == {0, 1}


def test_trainer_resume(tmp_path):
    args = ["--graph_type", "-1", "--master_port", "29813",
            "--overwrite_checkpoints", "True"]
    _run_trainer(tmp_path, args)
    # resume run continues from epoch 1 -> immediately exits (num_epochs=1)
    _run_trainer(tmp_path, args + ["--resume", "True"])


def test_trainer_mid_epoch_resume(tmp_path):
    """Mid-epoch resume fast-forwards the sampler (gossip_sgd.py
    train_epoch start_itr spoof; reference gossip_sgd.py:356-364):
    after resuming from a checkpoint with itr=1, iteration 0 must NOT
    be re-run and iteration 1 must be."""
    import pandas as pd
    import torch

    args = ["--graph_type", "-1", "--master_port", "29817",
            "--overwrite_checkpoints", "True", "--print_freq", "1"]
    _run_trainer(tmp_path, args)

    ckpt = f"{tmp_path}/ckpt/checkpoint_r0_n1.pth.tar"
    state = torch.load(ckpt, weights_only=False)
    assert state["epoch"] == 1 and state["itr"] == 0
    state["epoch"] = 0
    state["itr"] = 1
    torch.save(state, ckpt)

    _run_trainer(tmp_path, args + ["--resume", "True"])

    csv = f"{tmp_path}/ckpt/out_r0_n1.csv"
    # CSV preamble: BEGIN-TRAINING + 3 key,value rows before the header
    df = pd.read_csv(csv, skiprows=4)
    train = df[(df["itr"] >= 0) & (df["Epoch"] == 0)]
    n0 = len(train[train["itr"] == 0])
    n1 = len(train[train["itr"] == 1])
    # iteration 0 was NOT re-run (fast-forwarded); iteration 1 ran in
    # both the original pass and the resumed pass
    assert n0 == 1, (n0, n1)
    assert n1 > n0 + 1, (n0, n1)


def test_adpsgd_cli_two_rank(tmp_path):
    _run_trainer(
        tmp_path,
        ["--graph_type", "1", "--master_port", "29814",
         "--train_fast", "True"],
        script="gossip_sgd_adpsgd.py", nprocs=2,
    )
    assert os.path.exists(f"{tmp_path}/ckpt/out_r0_n2.csv")


def test_bench_json_contract(tmp_path):
    """The driver depends on bench.py's JSON line: validate schema."""
    import json

    env = dict(os.environ)
    r = subprocess.run(
        [
            sys.executable, os.path.join(REPO, "bench.py"),
            "--device", "cpu", "--steps", "2", "--warmup", "1",
            "--batch-size", "2", "--model", "resnet18", "--dtype", "fp32",
        ],
        env=env, check=True, timeout=300, cwd=str(tmp_path),
        capture_output=True, text=True,
    )
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in d, key
    assert d["metric"] == "images/sec"
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["config"]["model"] == "resnet18"
    assert d["value"] > 0


@pytest.mark.parametrize("algo", ["sgp", "osgp"])
def test_bench_eight_rank_cpu(tmp_path, algo):
    """Multi-rank bench.py integration over gloo:
    the exact torchrun path the driver's SCALE run uses, at world_size 8
    on CPU — exercises graph-capture fallback, the gossip thread per
    rank, and the gossip_ms aggregation."""
    import json

    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "8",
            "--master-addr", "127.0.0.1", "--master-port", "29515",
            os.path.join(REPO, "bench.py"),
            "--gpus", "8", "--device", "cpu", "--steps", "2",
            "--warmup", "1", "--batch-size", "1", "--model", "resnet18",
            "--dtype", "fp32", "--algorithm", algo,
            "--gossip-dtype", "fp32",
        ],
        env=env, check=True, timeout=600, cwd=str(tmp_path),
        capture_output=True, text=True,
    )
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == f"{algo}-dp8"
    assert d["value"] > 0
    # 8 ranks actually gossiped: the per-step gossip meter saw work
    assert d["config"]["gossip_ms_per_step"] > 0
