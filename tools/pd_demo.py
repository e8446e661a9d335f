"""PD-disaggregation demo/benchmark: 1 prefiller + 1 decoder over RCCL/xGMI.

BASELINE config #3 shape (Qwen3-8B PD-split on 2 MI355X). Launch:

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 tools/pd_demo.py --requests 32

Rank 0 = prefiller (kv_producer): prefills each prompt, samples the first
token, packs the request's KV blocks across all layers into one contiguous
staging tensor and ships it with a single RCCL send over an xGMI link.
Rank 1 = decoder (kv_consumer): receives + scatters KV, decodes to
completion with continuous batching. Works on CPU/gloo for smoke testing
(tiny model) and on 2 GPUs with RCCL.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="Qwen3-8B")
    p.add_argument("--requests", type=int, default=32)
    p.add_argument("--prompt-len", type=int, default=1024)
    p.add_argument("--gen-len", type=int, default=128)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    use_cuda = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    device = f"cuda:{local_rank}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local_rank)
    dist.init_process_group("nccl" if use_cuda else "gloo")

    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.distributed.kv_transfer import (
        KV_CONSUMER, KV_PRODUCER, RcclKVConnector,
    )
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config
    from fusioninfer_amd.pd import PDDecoder, PDPrefiller

    mc = get_model_config(args.model)
    cfg = EngineConfig(
        model=mc,
        cache=CacheConfig(num_gpu_blocks=None if use_cuda else 2048),
        scheduler=SchedulerConfig(
            max_num_seqs=64,
            max_num_batched_tokens=8192,
            max_model_len=args.prompt_len + args.gen_len + 64,
        ),
        enforce_eager=not use_cuda,
    )
    engine = LLMEngine(cfg, device=device)
    torch.manual_seed(7 + 0)  # prompts identical on both ranks
    prompts = [
        torch.randint(0, mc.vocab_size, (args.prompt_len,)).tolist()
        for _ in range(args.requests)
    ]

    dist.barrier()
    t0 = time.monotonic()
    if rank == 0:
        pre = PDPrefiller(engine, RcclKVConnector(KV_PRODUCER, 1, device=device))
        for prompt in prompts:
            pre.process(prompt)
        dist.barrier()
        if True:
            elapsed = time.monotonic() - t0
            print(json.dumps({
                "role": "prefiller", "requests": args.requests,
                "elapsed_s": round(elapsed, 3),
            }))
    else:
        dec = PDDecoder(engine, RcclKVConnector(KV_CONSUMER, 0, device=device))
        ids = [
            dec.accept(SamplingParams(max_tokens=args.gen_len))
            for _ in range(args.requests)
        ]
        results = dec.decode_all()
        dist.barrier()
        elapsed = time.monotonic() - t0
        total_tokens = sum(len(results[i].output_token_ids) for i in ids)
        print(json.dumps({
            "role": "decoder",
            "requests": args.requests,
            "elapsed_s": round(elapsed, 3),
            "goodput_req_per_s": round(args.requests / elapsed, 3),
            "gen_tokens_per_s": round(total_tokens / elapsed, 1),
        }))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
