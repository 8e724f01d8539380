"""End-to-end CLI test: tiny synthetic training run through main()."""

import json
import os

import pytest


def test_main_synthetic_run(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    from luminaai_amd.main import main
    result = main([
        "--preset", "debug", "--synthetic-steps", "3",
        "--experiment-name", "cli_test", "--precision", "fp32",
        "--micro-batch", "2", "--accum", "1", "--seq-len", "32",
        "--set", "hidden_size=64", "--set", "num_layers=2",
        "--set", "num_heads=4", "--set", "num_kv_heads=2",
        "--set", "vocab_size=512", "--set", "intermediate_size=128",
        "--set", "num_workers=0", "--set", "use_moe=false",
        "--set", "use_mod=false", "--set", "gradient_checkpointing=false",
        "--set", "eval_every_n_batches=0", "--set", "save_every_n_batches=0",
    ])
    assert result["global_step"] >= 3
    exp = tmp_path / "experiments" / "cli_test"
    assert (exp / "config.yaml").exists()
    assert (exp / "training_summary.json").exists()
    summary = json.loads((exp / "training_summary.json").read_text())
    assert summary["global_step"] >= 3
    assert (exp / "training_report.html").exists()


def test_main_data_run(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    from luminaai_amd.utils import generate_sample_data
    data = str(tmp_path / "train.jsonl")
    generate_sample_data(data, n=8)
    from luminaai_amd.main import main
    result = main([
        "--preset", "debug", "--train-data", data,
        "--experiment-name", "cli_data_test", "--precision", "fp32",
        "--epochs", "1", "--micro-batch", "2", "--accum", "1",
        "--seq-len", "32",
        "--set", "hidden_size=64", "--set", "num_layers=2",
        "--set", "num_heads=4", "--set", "num_kv_heads=2",
        "--set", "vocab_size=512", "--set", "intermediate_size=128",
        "--set", "num_workers=0", "--set", "use_moe=false",
        "--set", "use_mod=false", "--set", "gradient_checkpointing=false",
        "--set", "eval_every_n_batches=0", "--set", "save_every_n_batches=0",
        "--set", "eval_data_path=",
    ])
    assert result["global_step"] > 0


def test_unknown_config_field_rejected():
    from luminaai_amd.main import main
    with pytest.raises(SystemExit):
        main(["--preset", "debug", "--set", "not_a_field=1",
              "--synthetic-steps", "1"])


def test_main_zero3_synthetic(tmp_path, monkeypatch):
    """ZeRO-3 engine reachable from the CLI (single process)."""
    monkeypatch.chdir(tmp_path)
    from luminaai_amd.main import main
    result = main([
        "--preset", "debug", "--synthetic-steps", "2",
        "--experiment-name", "cli_z3", "--precision", "fp32",
        "--micro-batch", "2", "--accum", "1", "--seq-len", "32",
        "--zero", "3",
        "--set", "hidden_size=64", "--set", "num_layers=2",
        "--set", "num_heads=4", "--set", "num_kv_heads=2",
        "--set", "vocab_size=512", "--set", "intermediate_size=128",
        "--set", "num_workers=0", "--set", "use_moe=false",
        "--set", "use_mod=false", "--set", "gradient_checkpointing=false",
        "--set", "eval_every_n_batches=0", "--set", "save_every_n_batches=0",
    ])
    assert result["global_step"] >= 2


def test_main_eval_only(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    from luminaai_amd.utils import generate_sample_data
    data = str(tmp_path / "eval.jsonl")
    generate_sample_data(data, n=6)
    from luminaai_amd.main import main
    result = main([
        "--preset", "debug", "--train-data", data, "--eval-only",
        "--experiment-name", "cli_eval", "--precision", "fp32",
        "--micro-batch", "2", "--seq-len", "32",
        "--set", "hidden_size=64", "--set", "num_layers=2",
        "--set", "num_heads=4", "--set", "num_kv_heads=2",
        "--set", "vocab_size=512", "--set", "intermediate_size=128",
        "--set", "num_workers=0", "--set", "use_moe=false",
        "--set", "use_mod=false", "--set", "eval_data_path=",
    ])
    assert "loss" in result["eval"]
    assert result["eval"]["batches"] > 0


def _pp_cli_worker(rank, world, port, q, tmp):
    import traceback
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
    })
    os.chdir(tmp)
    try:
        from luminaai_amd.main import main
        result = main([
            "--preset", "debug", "--synthetic-steps", "2",
            "--experiment-name", "cli_pp", "--precision", "fp32",
            "--micro-batch", "2", "--accum", "2", "--seq-len", "32",
            "--pp", "2",
            "--set", "hidden_size=64", "--set", "num_layers=4",
            "--set", "num_heads=4", "--set", "num_kv_heads=2",
            "--set", "vocab_size=512", "--set", "intermediate_size=128",
            "--set", "num_workers=0", "--set", "use_moe=false",
            "--set", "use_mod=false", "--set", "gradient_checkpointing=false",
            "--set", "eval_every_n_batches=0", "--set", "save_every_n_batches=0",
        ])
        q.put((rank, "ok", result))
    except Exception:  # noqa: BLE001
        q.put((rank, "err", traceback.format_exc()))


def test_main_pipeline_parallel(tmp_path):
    """--pp 2 runs the dedicated pipeline loop end-to-end (gloo x2)."""
    import socket

    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    procs = [ctx.Process(target=_pp_cli_worker,
                         args=(r, 2, port, q, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=120)
    for r in range(2):
        assert results[r]["global_step"] >= 2
        assert (tmp_path / results[r]["checkpoint"]).exists()


def _sp_ring_cli_worker(rank, world, port, q, tmp):
    import traceback
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
    })
    os.chdir(tmp)
    try:
        from luminaai_amd.main import main
        result = main([
            "--preset", "debug", "--synthetic-steps", "2",
            "--experiment-name", "cli_spring", "--precision", "fp32",
            "--micro-batch", "2", "--accum", "1", "--seq-len", "64",
            "--sp", "2", "--sp-mode", "ring",
            "--set", "hidden_size=64", "--set", "num_layers=2",
            "--set", "num_heads=4", "--set", "num_kv_heads=2",
            "--set", "vocab_size=512", "--set", "intermediate_size=128",
            "--set", "num_workers=0", "--set", "use_moe=false",
            "--set", "use_mod=false", "--set", "gradient_checkpointing=false",
            "--set", "eval_every_n_batches=0", "--set", "save_every_n_batches=0",
        ])
        q.put((rank, "ok", result))
    except Exception:  # noqa: BLE001
        q.put((rank, "err", traceback.format_exc()))


def test_main_ring_sp(tmp_path):
    """--sp 2 --sp-mode ring trains end-to-end (gloo x2)."""
    import socket

    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    procs = [ctx.Process(target=_sp_ring_cli_worker,
                         args=(r, 2, port, q, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=120)
    assert results[0]["global_step"] >= 2


def _tp_cli_worker(rank, world, port, q, tmp):
    import traceback
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
    })
    os.chdir(tmp)
    try:
        from luminaai_amd.main import main
        result = main([
            "--preset", "debug", "--synthetic-steps", "2",
            "--experiment-name", "cli_tp", "--precision", "fp32",
            "--micro-batch", "2", "--accum", "1", "--seq-len", "64",
            "--tp", "2",
            "--set", "hidden_size=64", "--set", "num_layers=2",
            "--set", "num_heads=4", "--set", "num_kv_heads=2",
            "--set", "vocab_size=512", "--set", "intermediate_size=128",
            "--set", "num_workers=0", "--set", "use_moe=false",
            "--set", "use_mod=false", "--set", "gradient_checkpointing=false",
            "--set", "eval_every_n_batches=0", "--set", "save_every_n_batches=0",
        ])
        q.put((rank, "ok", result))
    except Exception:  # noqa: BLE001
        q.put((rank, "err", traceback.format_exc()))


def test_main_tensor_parallel(tmp_path):
    """--tp 2 trains end-to-end through the CLI (gloo x2)."""
    import socket

    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    procs = [ctx.Process(target=_tp_cli_worker,
                         args=(r, 2, port, q, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=120)
    for r in range(2):
        assert results[r]["global_step"] >= 2
