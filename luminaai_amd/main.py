"""Training entry point (CLI).

Rebuild of /root/reference/Src/Main_Scripts/Main.py:1506-3052 as a proper
argparse CLI (the reference configured training through a ~200-line in-file
dict, Main.py:1514-2007, with no flag parser). The 17-step startup becomes:
validate env -> config preset + overrides -> experiment dir -> tokenizer ->
datasets -> distributed init (+EP mesh) -> orchestrator (builds model+trainer)
-> chinchilla auto-epochs -> checkpoint resume -> signal handlers ->
OOM-protected adaptive training -> reports.

Reference quirks deliberately fixed: the orchestrator trains the SAME model
Main built (not a silently rebuilt second one, orchestrator.py:1153-1170);
initialize_training runs once (not twice, Main.py:2482,2502); use_mod reaches
the model config (Main.py:572-602 dropped it).
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import sys
import time
from typing import Optional

import torch

from .config import Config, ConfigPresets
from .data.dataset import SyntheticDataset, setup_datasets
from .data.tokenizer import ConversationTokenizer
from .models import DeepSeekTransformer, config_to_deepseek_config
from .monitoring import ProductionLogger, TrainingHealthMonitor, WandbLogger
from .parallel import comm
from .parallel.mesh import init_mesh
from .training import (AdaptiveTrainingOrchestrator, EnhancedChinchillaScaler)
from .utils import create_training_report, validate_environment


def build_arg_parser() -> argparse.ArgumentParser:
    ap = argparse.ArgumentParser(
        prog="luminaai-amd train",
        description="MI355X-native sparse-transformer training")
    ap.add_argument("--preset", default="debug",
                    help=f"config preset ({', '.join(ConfigPresets.names())})")
    ap.add_argument("--config", default=None, help="YAML config file (overrides preset)")
    ap.add_argument("--train-data", default=None)
    ap.add_argument("--eval-data", default=None)
    ap.add_argument("--synthetic-steps", type=int, default=None,
                    help="train on synthetic data for N optimizer steps "
                         "(no data files needed)")
    ap.add_argument("--experiment-name", default=None)
    ap.add_argument("--resume", default=None,
                    help="checkpoint path, or 'latest'/'best'")
    ap.add_argument("--epochs", type=int, default=None)
    ap.add_argument("--lr", type=float, default=None)
    ap.add_argument("--micro-batch", type=int, default=None)
    ap.add_argument("--accum", type=int, default=None)
    ap.add_argument("--seq-len", type=int, default=None)
    ap.add_argument("--zero", type=int, default=None, choices=(0, 1, 2, 3))
    ap.add_argument("--ep", type=int, default=None,
                    help="expert-parallel degree")
    ap.add_argument("--pp", type=int, default=1,
                    help="pipeline-parallel stages (= world size; dedicated "
                         "1F1B loop, see training/pipeline_loop.py)")
    ap.add_argument("--pp-virtual", type=int, default=1,
                    help="virtual stage chunks per PP rank (interleaved "
                         "schedule)")
    ap.add_argument("--tp", type=int, default=1,
                    help="tensor-parallel degree (dense blocks)")
    ap.add_argument("--sp-mode", default="ulysses",
                    choices=["ulysses", "ring"],
                    help="sequence-parallel exchange: head<->seq all-to-all "
                         "or ring attention (no head-count limit)")
    ap.add_argument("--sp", type=int, default=1,
                    help="Ulysses sequence-parallel degree")
    ap.add_argument("--precision", default=None,
                    choices=("auto", "fp32", "bf16", "fp16", "fp8"))
    ap.add_argument("--no-adaptive", action="store_true",
                    help="disable the adaptive orchestrator interventions")
    ap.add_argument("--chinchilla", action="store_true",
                    help="auto-set epochs from Chinchilla scaling")
    ap.add_argument("--eval-only", action="store_true",
                    help="evaluate --eval-data (or train data) with the "
                         "resumed checkpoint and exit")
    ap.add_argument("--set", action="append", default=[], metavar="KEY=VALUE",
                    help="override any Config field")
    return ap


def apply_overrides(cfg: Config, args) -> Config:
    simple = {
        "train_data": "train_data_path", "eval_data": "eval_data_path",
        "experiment_name": "experiment_name", "epochs": "num_epochs",
        "lr": "learning_rate", "micro_batch": "micro_batch_size",
        "accum": "gradient_accumulation_steps", "seq_len": "seq_length",
        "zero": "zero_stage", "precision": "precision",
        "ep": "expert_parallel_size",
    }
    for arg_name, field in simple.items():
        v = getattr(args, arg_name)
        if v is not None:
            setattr(cfg, field, v)
    if args.no_adaptive:
        cfg.enable_adaptive_lr = False
    for kv in args.set:
        key, _, val = kv.partition("=")
        if not hasattr(cfg, key):
            raise SystemExit(f"unknown config field {key!r}")
        cur = getattr(cfg, key)
        if isinstance(cur, bool):
            val = val.lower() in ("1", "true", "yes")
        elif isinstance(cur, int):
            val = int(val)
        elif isinstance(cur, float):
            val = float(val)
        setattr(cfg, key, val)
    return cfg


def main(argv: Optional[list] = None) -> dict:
    args = build_arg_parser().parse_args(argv)
    cfg = Config.load(args.config) if args.config \
        else ConfigPresets.get(args.preset)
    cfg = apply_overrides(cfg, args)
    cfg.experiment_name = cfg.experiment_name or \
        f"{args.preset}_{time.strftime('%Y%m%d_%H%M%S')}"
    cfg.validate()

    distributed = comm.init_distributed()
    rank = comm.get_rank()
    world = comm.get_world_size()
    ep = cfg.expert_parallel_size or 1
    if args.sp > 1 or args.tp > 1:
        ep = 1        # SP/TP claim the mesh; auto-EP must stand down
    elif world > 1 and cfg.use_moe and ep <= 1 \
            and cfg.num_experts % world == 0:
        ep = world
    if world > 1:
        mesh = init_mesh(ep, sp_size=args.sp, tp_size=args.tp,
                         sp_mode=args.sp_mode)
    else:
        mesh = init_mesh(1)

    exp_dir = os.path.join("experiments", cfg.experiment_name)
    logger = ProductionLogger("luminaai", log_dir=exp_dir if rank == 0 else None)
    if rank == 0:
        os.makedirs(exp_dir, exist_ok=True)
        cfg.save(os.path.join(exp_dir, "config.yaml"))
        env = validate_environment(cfg)
        for w in env["warnings"]:
            logger.warning(w)
        for e in env["errors"]:
            logger.error(e)
        if not env["ok"]:
            logger.error("environment validation failed — continuing anyway")

    torch.manual_seed(cfg.seed + rank)
    tokenizer = ConversationTokenizer(max_length=cfg.seq_length,
                                      assistant_loss_weight=cfg.assistant_loss_weight)

    # datasets
    if args.synthetic_steps:
        n = (cfg.micro_batch_size or 1) * cfg.gradient_accumulation_steps \
            * args.synthetic_steps
        # PP: every stage of one pipeline column MUST see the same stream
        # (rank 0 consumes input_ids, the last stage consumes the matching
        # labels) — seed per DP replica, not per rank
        seed = cfg.seed + (rank // args.pp if args.pp > 1 else rank)
        train_ds = SyntheticDataset(cfg.vocab_size, cfg.seq_length, n,
                                    seed=seed)
        eval_ds = None
        cfg.num_epochs = 1
    else:
        train_ds, eval_ds = setup_datasets(cfg, tokenizer)
        if train_ds is None:
            logger.error(f"no training data at {cfg.train_data_path}; "
                         "pass --train-data or --synthetic-steps")
            raise SystemExit(2)

    if cfg.compile and rank == 0:
        logger.warning("config.compile requested but torch.compile is "
                       "disabled by design on this framework (its ROCm "
                       "backend is Triton); the native path is HIP kernels "
                       "+ hipGraphs")
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    if args.pp > 1:
        if world % args.pp != 0:
            raise SystemExit(f"--pp {args.pp} requires world size divisible "
                             f"by {args.pp} (got {world}); ranks lay out as "
                             "dp_replica x pp_stage")
        from .training.pipeline_loop import run_pipeline_training
        result = run_pipeline_training(
            model, cfg, train_ds, logger, pp=args.pp,
            virtual_stages=max(1, args.pp_virtual),
            steps=args.synthetic_steps)
        if rank == world - 1:
            logger.info(f"pipeline training done: step "
                        f"{result['global_step']} loss {result['loss']:.4f}")
        return result
    if mesh.tp_size > 1:
        from .parallel.tensor_parallel import convert_to_tensor_parallel
        n_tp = convert_to_tensor_parallel(model, mesh)
        logger.info(f"tensor parallel: {n_tp} layers sharded {mesh.tp_size}-way")
    if rank == 0:
        fp = model.get_memory_footprint()
        logger.info(f"model: {fp['total_params'] / 1e6:.1f}M total / "
                    f"{fp['active_params'] / 1e6:.1f}M active params")

    orch = AdaptiveTrainingOrchestrator(cfg, model=model, tokenizer=tokenizer,
                                        logger=logger)
    trainer = orch.initialize_training()

    # Chinchilla auto-epochs
    if args.chinchilla and hasattr(train_ds, "__len__"):
        scaler = EnhancedChinchillaScaler(cfg, model=model)
        ds_tokens = len(train_ds) * cfg.seq_length
        cfg.num_epochs = scaler.compute_optimal_epochs(ds_tokens)
        logger.info(f"chinchilla: {scaler.optimal_tokens:.2e} optimal tokens "
                    f"-> {cfg.num_epochs} epochs")

    # resume
    if args.resume:
        try:
            payload = trainer.load_checkpoint(args.resume)
            logger.info(f"resumed from step {payload.get('global_step')}")
        except FileNotFoundError:
            logger.warning(f"no checkpoint {args.resume!r}; starting fresh")

    # emergency save on SIGINT/SIGTERM (reference Main.py:1126-1152)
    def _emergency(sig, frame):
        logger.warning(f"signal {sig}: emergency checkpoint")
        try:
            if rank == 0:
                trainer.checkpoints.emergency_save(
                    trainer.model, global_step=trainer.global_step)
        finally:
            orch.cleanup()
            sys.exit(128 + sig)

    if rank == 0 and threading_main():
        signal.signal(signal.SIGINT, _emergency)
        signal.signal(signal.SIGTERM, _emergency)

    health = TrainingHealthMonitor(check_every=cfg.health_check_interval)
    wb = WandbLogger(cfg, enabled=cfg.enable_wandb and rank == 0)
    prom = None
    if cfg.prometheus_port and rank == 0:
        try:
            from .monitoring.prometheus import PrometheusExporter
            prom = PrometheusExporter(port=cfg.prometheus_port)
            logger.info(f"prometheus scrape endpoint on :{cfg.prometheus_port}")
        except Exception as e:  # noqa: BLE001
            logger.warning(f"prometheus exporter unavailable: {e}")

    def _hook(m):
        orch._enqueue_metrics(m)
        if prom is not None:
            prom(m)
        floats = {k: v for k, v in m.as_dict().items()
                  if isinstance(v, (int, float))}
        health.log_step(floats, m.step)
        wb.log(floats, m.step)

    trainer.set_metrics_hook(_hook)

    t0 = time.time()
    if args.eval_only:
        from .data.dataset import create_dataloader
        ds = eval_ds if eval_ds is not None else train_ds
        dl = create_dataloader(ds, cfg, shuffle=False)
        stats = trainer.evaluate(dl)
        if rank == 0:
            print(json.dumps({"eval": stats}, default=str))
        orch.cleanup()
        wb.finish()
        comm.cleanup()
        return {"eval": stats, "global_step": trainer.global_step}
    try:
        result = run_with_oom_protection(orch, trainer, cfg, logger,
                                         train_ds, eval_ds)
    finally:
        orch.cleanup()
        wb.finish()
        comm.cleanup()

    if rank == 0:
        summary = {
            "experiment": cfg.experiment_name,
            "wall_seconds": time.time() - t0,
            "global_step": trainer.global_step,
            "final_metrics": trainer._metric_floats(),
            "interventions": orch.interventions_executed,
            "health": health.health_check(),
        }
        with open(os.path.join(exp_dir, "training_summary.json"), "w") as f:
            json.dump(summary, f, indent=2, default=str)
        create_training_report(result.get("epochs", []), cfg,
                               os.path.join(exp_dir, "training_report.html"),
                               extra=summary)
        health.save_report(os.path.join(exp_dir, "health_report.json"))
        logger.info(f"done: {summary['global_step']} steps in "
                    f"{summary['wall_seconds']:.0f}s -> {exp_dir}")
    return result


def run_with_oom_protection(orch, trainer, cfg, logger, train_ds, eval_ds,
                            max_attempts: int = 10):
    """Halve micro-batch / double accumulation and retry on OOM
    (reference Main.py:292-501)."""
    for attempt in range(max_attempts):
        try:
            return orch.run_adaptive_training(train_ds, eval_ds)
        except torch.cuda.OutOfMemoryError:
            torch.cuda.empty_cache()
            mb = cfg.micro_batch_size or 1
            if mb > 1:
                cfg.micro_batch_size = mb // 2
            else:
                cfg.gradient_accumulation_steps *= 2
                trainer.accum_steps = cfg.gradient_accumulation_steps
            logger.warning(
                f"OOM (attempt {attempt + 1}): micro_batch="
                f"{cfg.micro_batch_size} accum={cfg.gradient_accumulation_steps}")
    raise RuntimeError(f"training failed after {max_attempts} OOM retries")


def threading_main() -> bool:
    import threading
    return threading.current_thread() is threading.main_thread()


if __name__ == "__main__":
    main()
