"""flatten/unflatten round-trips, Meter stats, is_power_of."""

import math
import os

import pytest
import torch

from stochastic_gradient_push_amd.utils import (
    Meter,
    flatten_tensors,
    group_by_dtype,
    is_power_of,
    unflatten_tensors,
)


def test_flatten_unflatten_roundtrip():
    tensors = [torch.randn(3, 4), torch.randn(7), torch.randn(2, 2, 2)]
    flat = flatten_tensors(tensors)
    assert flat.numel() == sum(t.numel() for t in tensors)
    out = unflatten_tensors(flat, tensors)
    for a, b in zip(tensors, out):
        assert torch.equal(a, b)


def test_flatten_single():
    t = [torch.randn(5, 5)]
    flat = flatten_tensors(t)
    assert flat.shape == (25,)
    # must be a copy, not a view
    flat.add_(1)
    assert not torch.allclose(flat.view(5, 5), t[0])


def test_group_by_dtype():
    ts = [torch.randn(2), torch.randn(2).double(), torch.randn(3)]
    g = group_by_dtype(ts)
    assert len(g[torch.float32]) == 2
    assert len(g[torch.float64]) == 1


def test_meter_stats():
    m = Meter(stateful=True)
    vals = [1.0, 2.0, 3.0, 4.0]
    for v in vals:
        m.update(v)
    assert m.avg == pytest.approx(2.5)
    mean = sum(vals) / 4
    std = (sum((v - mean) ** 2 for v in vals) / 3) ** 0.5
    assert m.std == pytest.approx(std)
    assert m.mad == pytest.approx(1.0)
    assert str(m) == "4.000,2.500,1.000"


def test_meter_csv_and_pretty():
    m = Meter(ptag="BT", stateful=False, csv_format=False)
    m.update(1.5)
    assert str(m).startswith("BT: 1.500")


def test_meter_state_dict_roundtrip():
    m = Meter(stateful=True)
    for v in (1.0, 5.0):
        m.update(v)
    m2 = Meter(init_dict=m.state_dict(), stateful=True)
    assert m2.avg == m.avg and m2.count == m.count


def test_is_power_of():
    assert is_power_of(8, 2)
    assert is_power_of(27, 3)
    assert not is_power_of(12, 2)
    assert is_power_of(1, 0)
    assert not is_power_of(5, 1)


def test_transformer_log_parser(tmp_path):
    from visualization.plotting import parse_transformer_out

    log = tmp_path / "train.log"
    log.write_text(
        "| epoch 001 | valid on 'valid' subset | nll_loss 9.123 "
        "| num_updates 100\n"
        "| epoch 002 | valid on 'valid' subset | nll_loss 7.5 "
        "| num_updates 200\n"
        "| training line without valid keyword | loss 3\n"
    )
    df = parse_transformer_out(str(log))
    assert list(df["num_updates"]) == [100, 200]
    assert df["valid_nll_loss"].iloc[1] == 7.5


def test_plotting_renders_png(tmp_path):
    """plot_itrs/plot_scaling produce image files from a real run's CSV."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update(RANK="0", WORLD_SIZE="1", MASTER_ADDR="127.0.0.1")
    subprocess.run(
        [
            sys.executable, os.path.join(repo, "gossip_sgd.py"),
            "--num_epochs", "1", "--num_iterations_per_training_epoch", "4",
            "--batch_size", "2", "--synthetic_size", "16",
            "--model", "resnet18", "--num_classes", "10",
            "--image_size", "32", "--num_dataloader_workers", "0",
            "--device", "cpu", "--checkpoint_dir", f"{tmp_path}/ck/",
            "--num_itr_ignore", "0", "--graph_type", "-1",
            "--train_fast", "True", "--print_freq", "1",
            "--master_port", "29821",
        ],
        env=env, check=True, timeout=240, cwd=str(tmp_path),
    )
    from visualization.plotting import load_experiment, plot_itrs, plot_scaling

    runs = load_experiment(f"{tmp_path}/ck/")
    out = plot_itrs(runs, f"{tmp_path}/loss.png")
    assert os.path.exists(out)
    out2 = plot_scaling({1: 0.5, 2: 0.3, 4: 0.2}, f"{tmp_path}/scale.png")
    assert os.path.exists(out2)


def test_experiment_matrix_and_plot(tmp_path):
    """ETH/IB experiment-tag matrix (reference plotting.py:55-134
    equivalent): matrix shape is consistent and plot_matrix renders,
    skipping absent experiments and including present ones."""
    import shutil
    import subprocess
    import sys

    from visualization.plotting import experiment_matrix, plot_matrix

    for kind, n_groups in (("eth", 3), ("ib", 2), ("transformer", 2)):
        nodes, fpaths, tags, legends, colors = experiment_matrix(kind)
        assert len(fpaths) == len(tags) == len(legends) == len(colors) \
            == n_groups
        for t, l, c in zip(tags, legends, colors):
            assert len(t) == len(l) == len(c) == len(nodes)

    # produce one real run and place it under an ETH tag
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update(RANK="0", WORLD_SIZE="1", MASTER_ADDR="127.0.0.1")
    subprocess.run(
        [
            sys.executable, os.path.join(repo, "gossip_sgd.py"),
            "--num_epochs", "1", "--num_iterations_per_training_epoch", "4",
            "--batch_size", "2", "--synthetic_size", "16",
            "--model", "resnet18", "--num_classes", "10",
            "--image_size", "32", "--num_dataloader_workers", "0",
            "--device", "cpu", "--checkpoint_dir", f"{tmp_path}/ck/",
            "--num_itr_ignore", "0", "--graph_type", "-1",
            "--train_fast", "True", "--print_freq", "1",
            "--master_port", "29822",
        ],
        env=env, check=True, timeout=240, cwd=str(tmp_path),
    )
    res = tmp_path / "results_dir" / "out_files"
    res.mkdir(parents=True)
    shutil.copy(
        f"{tmp_path}/ck/out_r0_n1.csv", res / "SGP-4ETHout_r0_n4.csv"
    )
    out = plot_matrix(
        "eth", f"{tmp_path}/eth.png",
        results_dir=str(tmp_path / "results_dir"), metric="avg:Loss",
    )
    assert os.path.exists(out)


def test_example_distributed_averaging(tmp_path):
    """The standalone averaging example runs end to end (4 gloo ranks,
    reference README.md:67-68 usable-standalone contract)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    r = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "4",
            "--master-addr", "127.0.0.1", "--master-port", "29531",
            os.path.join(repo, "examples", "distributed_averaging.py"),
        ],
        env=env, timeout=300, cwd=str(tmp_path),
        capture_output=True, text=True,
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "converged" in r.stdout.lower() or "average" in r.stdout.lower()


def test_tools_compile():
    """Every tools/ script parses and the launch scripts are present
    (keeps utility drift out of the GPU-validated paths)."""
    import py_compile

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    tools = os.path.join(repo, "tools")
    names = sorted(os.listdir(tools))
    assert {"analyze_ktrace.py", "bringup_multigpu.py", "conv_bench.py",
            "gemm_bench.py", "kernels_pmc.py"} <= set(names)
    for n in names:
        if n.endswith(".py"):
            py_compile.compile(os.path.join(tools, n), doraise=True)
    scripts = os.listdir(os.path.join(repo, "job_scripts"))
    assert any(s.startswith("submit_SGP") for s in scripts)
    assert any("single_node" in s for s in scripts)


def test_nic_discovery_and_env_pinning(monkeypatch):
    """NIC autodiscovery + RCCL/Gloo env pinning (reference
    experiment_utils/helpers.py:44-67, gossip_sgd.py:654-666)."""
    from stochastic_gradient_push_amd.utils import nic

    # deterministic fake interface universe
    monkeypatch.setattr(
        nic.os, "listdir", lambda _: ["lo", "ib0", "ens3f0", "docker0"]
    )

    class FakeOut:
        stdout = b"2: ens3f0: <UP> ...\n3: ib0: <UP> ...\n"

    monkeypatch.setattr(nic.subprocess, "run",
                        lambda *a, **k: FakeOut())
    assert nic.get_tcp_interface_name("ethernet") == "ens3f0"
    assert nic.get_tcp_interface_name("infiniband") == "ib0"

    # pin_comm_env mutates the real process env; the fake interface
    # name must NOT leak into later tests (gloo would try to bind to
    # it), so pop everything this block touches
    try:
        nic.pin_comm_env("gloo", "ethernet")
        assert nic.os.environ["GLOO_SOCKET_IFNAME"] == "ens3f0"
        nic.pin_comm_env("nccl", "ethernet")
        assert nic.os.environ["NCCL_SOCKET_IFNAME"] == "ens3f0"
        assert nic.os.environ["NCCL_IB_DISABLE"] == "1"
    finally:
        for var in ("GLOO_SOCKET_IFNAME", "NCCL_SOCKET_IFNAME",
                    "NCCL_IB_DISABLE"):
            nic.os.environ.pop(var, None)

    # interface of the requested type absent -> loud failure
    monkeypatch.setattr(nic.os, "listdir", lambda _: ["lo", "docker0"])
    with pytest.raises(RuntimeError):
        nic.get_tcp_interface_name("infiniband")
