"""Engine server entrypoint: `python -m fusioninfer_amd.server ...`

CLI surface mirrors the engine flags the reference controller passes via
container args (--model / --tensor-parallel-size /
--kv-transfer-config '{"kv_connector":...,"kv_role":...}' / --max-model-len
/ port 8000), plus the rendezvous flags the control plane's LWS wrapper
appends for multi-node roles (--nnodes/--node-rank/--nproc-per-node/
--master-addr/--master-port, consuming LWS_LEADER_ADDRESS).
"""

from __future__ import annotations

import argparse
import json
import os


def parse_args(argv=None):
    p = argparse.ArgumentParser("fusioninfer-amd engine server")
    p.add_argument("--model", default="Qwen3-8B")
    p.add_argument("--model-path", default=None,
                   help="HF-layout safetensors dir; default random init")
    p.add_argument("--tokenizer", default=None,
                   help="path to a vendored tokenizer.json (or a dir "
                        "containing one); default: the checkpoint dir's, "
                        "else the reversible byte-level fallback")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--tensor-parallel-size", type=int, default=1)
    p.add_argument("--pipeline-parallel-size", type=int, default=1)
    p.add_argument("--data-parallel-size", type=int, default=1)
    p.add_argument("--max-model-len", type=int, default=8192)
    p.add_argument("--max-num-seqs", type=int, default=256)
    p.add_argument("--max-num-batched-tokens", type=int, default=8192)
    p.add_argument("--gpu-memory-utilization", type=float, default=0.85)
    p.add_argument("--kv-transfer-config", type=str, default=None,
                   help='JSON: {"kv_connector": "RcclConnector", '
                        '"kv_role": "kv_producer"|"kv_consumer"}')
    p.add_argument("--enforce-eager", action="store_true")
    p.add_argument("--quantization", choices=["fp8"], default=None)
    p.add_argument("--kv-cache-dtype", choices=["auto", "fp8"], default="auto")
    p.add_argument("--swap-space", type=float, default=0.0,
                   help="GiB of CPU swap for preempted sequences (0=recompute)")
    p.add_argument("--enable-lora", action="store_true")
    p.add_argument("--max-loras", type=int, default=8)
    p.add_argument("--max-cpu-loras", type=int, default=16,
                   help="accepted for reference-surface parity; adapters "
                        "are small enough to stay resident")
    p.add_argument("--lora-modules", nargs="*", default=[],
                   help="adapters to register at startup: name=/peft/dir "
                        "(loads adapter_model.safetensors) or name[=rank] "
                        "(seeded synthetic adapter)")
    p.add_argument("--enable-prefix-caching", action="store_true")
    p.add_argument("--scheduling-policy", choices=["fcfs", "priority"],
                   default="fcfs")
    p.add_argument("--speculative-config", type=str, default=None,
                   help='JSON: {"method": "ngram", '
                        '"num_speculative_tokens": 4, '
                        '"prompt_lookup_max": 4, "prompt_lookup_min": 2}')
    p.add_argument("--api-key", default=None,
                   help="bearer token required on every endpoint except "
                        "/health and /metrics (vLLM --api-key)")
    p.add_argument("--served-model-name", default=None,
                   help="name reported by /v1/models (vLLM parity)")
    # multi-node rendezvous flags injected by the LWS wrapper
    p.add_argument("--nnodes", type=int, default=1)
    p.add_argument("--node-rank", type=int, default=0)
    p.add_argument("--nproc-per-node", type=int, default=1)
    p.add_argument("--master-addr", default=None)
    p.add_argument("--master-port", type=int, default=29500)
    return p.parse_args(argv)


def build_engine_config(args):
    from fusioninfer_amd.config import (
        CacheConfig,
        EngineConfig,
        KVTransferConfig,
        ParallelConfig,
        SchedulerConfig,
    )
    from fusioninfer_amd.models.registry import get_model_config

    kvt = KVTransferConfig()
    if args.kv_transfer_config:
        raw = json.loads(args.kv_transfer_config)
        role = raw.get("kv_role")
        kvt = KVTransferConfig(
            kv_connector=raw.get("kv_connector"),
            kv_role=role,
            # convention: prefiller is rank 0 of the PD pair, decoder rank 1
            kv_rank=int(raw.get("kv_rank", 0 if role == "kv_producer" else 1)),
            kv_world_size=int(raw.get("kv_world_size", 2)),
        )
    mc = get_model_config(args.model)
    mc.model_path = args.model_path
    mc.quantization = args.quantization
    speculative = None
    if args.speculative_config:
        from fusioninfer_amd.engine.spec_decode import SpeculativeConfig

        raw = json.loads(args.speculative_config)
        speculative = SpeculativeConfig(
            method=raw.get("method", "ngram"),
            num_speculative_tokens=int(raw.get("num_speculative_tokens", 4)),
            prompt_lookup_max=int(raw.get("prompt_lookup_max", 4)),
            prompt_lookup_min=int(raw.get("prompt_lookup_min", 2)),
            disable_by_batch_size=int(
                raw.get("disable_by_batch_size", 32)
            ),
            model=raw.get("model"),
            draft_gpu_blocks=raw.get("draft_gpu_blocks"),
        )
    return EngineConfig(
        model=mc,
        cache=CacheConfig(
            gpu_memory_utilization=args.gpu_memory_utilization,
            enable_prefix_caching=args.enable_prefix_caching,
            kv_cache_dtype=args.kv_cache_dtype,
            swap_space_gb=args.swap_space,
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=args.max_num_seqs,
            max_num_batched_tokens=args.max_num_batched_tokens,
            max_model_len=args.max_model_len,
            policy=args.scheduling_policy,
        ),
        parallel=ParallelConfig(
            tensor_parallel_size=args.tensor_parallel_size,
            pipeline_parallel_size=args.pipeline_parallel_size,
            data_parallel_size=args.data_parallel_size,
        ),
        kv_transfer=kvt,
        speculative=speculative,
        max_loras=args.max_loras,
        max_cpu_loras=args.max_cpu_loras,
        enforce_eager=args.enforce_eager,
    )


def _rank_main(local_rank: int, args, nproc: int):
    """One engine rank. Global rank 0 serves HTTP; others execute the TP
    driver's broadcast batches (engine.worker_loop)."""
    import torch

    rank = args.node_rank * nproc + local_rank
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(local_rank)
    os.environ["WORLD_SIZE"] = str(args.nnodes * nproc)
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
    else:
        device = "cpu"

    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.server.api_server import build_app
    from fusioninfer_amd.server.serving import ServingEngine

    cfg = build_engine_config(args)
    tp_deg = cfg.parallel.tensor_parallel_size
    if rank != 0:
        engine = LLMEngine(cfg, device=device)
        if cfg.kv_transfer.kv_connector and tp_deg > 1:
            # PD x TP: worker ranks join their cross-side pair group and
            # ship/receive their own KV shard on pd_send/pd_recv commands
            from fusioninfer_amd.pd import build_pd_connector

            engine.kv_connector = build_pd_connector(
                cfg.kv_transfer, device, tp=tp_deg
            )
        engine.worker_loop()
        return
    kv_connector = None
    if cfg.kv_transfer.kv_connector:
        from fusioninfer_amd.pd import build_pd_connector

        if tp_deg > 1:
            # pair groups need the global world up first (world = 2*tp)
            from fusioninfer_amd.distributed import parallel_state as ps

            ps.init_distributed(
                tp_deg,
                backend="nccl" if device.startswith("cuda") else "gloo",
            )
        kv_connector = build_pd_connector(cfg.kv_transfer, device, tp=tp_deg)
    serving = ServingEngine(cfg, device=device, kv_connector=kv_connector)
    if args.enable_lora or args.lora_modules:
        for spec in args.lora_modules:
            name, _, val = spec.partition("=")
            if val and not val.isdigit():
                serving.engine.add_lora_from_path(name, val)
            else:
                serving.engine.add_lora(name, rank=int(val) if val else 16)

    import uvicorn

    from fusioninfer_amd.tokenizer import get_tokenizer

    tokenizer = get_tokenizer(
        cfg.model.vocab_size, args.model_path, args.tokenizer
    )
    app = build_app(serving, args.served_model_name or args.model,
                    tokenizer=tokenizer, api_key=args.api_key)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")
    serving.engine.stop_workers()


def main(argv=None):
    """Single- or multi-rank launch. For multi-node roles the control
    plane's LWS wrapper appends --nnodes/--node-rank/--nproc-per-node/
    --master-addr (from LWS_LEADER_ADDRESS) — this entrypoint then spawns
    one engine process per local GPU and forms the RCCL process group
    itself (the reference delegated this to Ray)."""
    args = parse_args(argv)
    if args.master_addr:
        os.environ["MASTER_ADDR"] = args.master_addr
        os.environ["MASTER_PORT"] = str(args.master_port)
    else:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", str(args.master_port))

    nproc = max(args.nproc_per_node, 1)
    world = args.nnodes * nproc
    if world <= 1:
        _rank_main(0, args, nproc)
        return
    assert args.tensor_parallel_size * args.pipeline_parallel_size == world, (
        "multi-rank launch maps every rank into one TP group or one "
        f"pipeline: tp*pp != {world}"
    )
    import torch.multiprocessing as mp

    procs = []
    ctx = mp.get_context("spawn")
    for lr in range(1, nproc):
        p = ctx.Process(target=_rank_main, args=(lr, args, nproc), daemon=True)
        p.start()
        procs.append(p)
    _rank_main(0, args, nproc)
    for p in procs:
        p.join(timeout=10)


if __name__ == "__main__":
    main()
