"""Multi-process distributed tests (gloo, world_size=2, CPU).

Covers the two distributed paths the MI355X build must get right by
construction (8-GPU runs are driver-side only):
 * PD disaggregation: prefiller -> decoder KV handoff produces the same
   tokens as a monolithic engine.
 * TP=2: sharded model logits match the TP=1 model (same seed => same
   logical weights, since shards are slices of the full matrices).
"""

import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.models.registry import get_model_config

PROMPT = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 8, 9, 7, 9, 3, 2, 3, 8, 4, 6, 2]
N_TOKENS = 6


def _engine_cfg(seed=11):
    return EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=64),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=512, max_model_len=128
        ),
        seed=seed,
    )


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)


# ------------------------------------------------------------------- PD
def _pd_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.distributed.kv_transfer import (
        KV_CONSUMER,
        KV_PRODUCER,
        RcclKVConnector,
    )
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.pd import PDDecoder, PDPrefiller

    _init(rank, 2, port)
    try:
        engine = LLMEngine(_engine_cfg(), device="cpu")
        if rank == 0:
            pre = PDPrefiller(engine, RcclKVConnector(KV_PRODUCER, peer_rank=1))
            first = pre.process(PROMPT)
            # monolithic reference on the same weights
            mono = LLMEngine(_engine_cfg(), device="cpu")
            outs = mono.generate(
                [PROMPT], SamplingParams(max_tokens=N_TOKENS, temperature=0.0)
            )
            expected = outs[0].output_token_ids
            assert expected[0] == first, (expected, first)
            exp = torch.tensor(expected, dtype=torch.long)
            got = torch.zeros(N_TOKENS, dtype=torch.long)
            dist.recv(got, src=1)
            assert torch.equal(exp, got), (exp, got)
            results[0] = "ok"
        else:
            dec = PDDecoder(engine, RcclKVConnector(KV_CONSUMER, peer_rank=0))
            req_id = dec.accept(SamplingParams(max_tokens=N_TOKENS, temperature=0.0))
            outs = dec.decode_all()
            toks = outs[req_id].output_token_ids
            assert len(toks) == N_TOKENS
            dist.send(torch.tensor(toks, dtype=torch.long), dst=0)
            results[1] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_pd_kv_handoff_matches_monolithic():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29611
        procs = [
            ctx.Process(target=_pd_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# ------------------------------------------------------------------- TP
def _tp_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.block_manager import BlockManager
    from fusioninfer_amd.engine.model_runner import ModelRunner
    from fusioninfer_amd.engine.sequence import SamplingParams, Sequence

    # TP=1 reference FIRST (before process-group init)
    ps.ensure_single_process()
    cfg = _engine_cfg()
    runner1 = ModelRunner(cfg, "cpu")
    runner1.allocate_kv_caches()
    bm1 = BlockManager(64, 16)
    seq = Sequence("s", PROMPT, SamplingParams())
    bm1.allocate(seq)
    ref_logits = runner1.execute_prefill([seq], bm1).float()
    ps.destroy()

    _init(rank, 2, port)
    try:
        ps.init_distributed(tensor_parallel_size=2, backend="gloo")
        cfg2 = _engine_cfg()
        cfg2.parallel.tensor_parallel_size = 2
        runner2 = ModelRunner(cfg2, "cpu")
        runner2.allocate_kv_caches()
        bm2 = BlockManager(64, 16)
        seq2 = Sequence("s2", PROMPT, SamplingParams())
        bm2.allocate(seq2)
        tp_logits = runner2.execute_prefill([seq2], bm2).float()
        rel = (tp_logits - ref_logits).norm() / ref_logits.norm()
        assert rel.item() < 0.05, rel.item()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_tp2_matches_tp1_logits():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29613
        procs = [
            ctx.Process(target=_tp_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# ------------------------------------------------------ TP engine (driver/worker)
def _tp_engine_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams

    _init(rank, 2, port)
    try:
        cfg = _engine_cfg()
        cfg.parallel.tensor_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        if engine.is_driver:
            outs = engine.generate(
                [PROMPT, [9, 9, 2] * 8],
                SamplingParams(max_tokens=4, temperature=0.0),
            )
            assert all(len(o.output_token_ids) == 4 for o in outs)
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_tp2_engine_driver_worker():
    """Full engine under TP=2: driver schedules + broadcasts, worker executes
    the same forwards (per-layer all-reduces line up); generation completes."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29617
        procs = [
            ctx.Process(target=_tp_engine_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# -------------------------------------------------- TP LoRA (rank-synced)
def _tp_lora_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams

    # TP=1 reference run first: base vs adapter outputs
    ps.ensure_single_process()
    torch.manual_seed(123)
    cfg1 = _engine_cfg()
    eng1 = LLMEngine(cfg1, device="cpu")
    eng1.add_lora("tuned", rank=4, seed=99)
    base1 = eng1.generate([PROMPT], SamplingParams(max_tokens=4,
                                                   temperature=0.0))[0]
    req = eng1.add_request(PROMPT, SamplingParams(max_tokens=4,
                                                  temperature=0.0),
                           lora_name="tuned")
    lora1 = None
    while eng1.has_unfinished():
        for o in eng1.step():
            if o.finished and o.request_id == req:
                lora1 = o
    ps.destroy()

    _init(rank, 2, port)
    try:
        torch.manual_seed(123)
        cfg = _engine_cfg()
        cfg.parallel.tensor_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        if engine.is_driver:
            # broadcasts the registration to the worker rank
            engine.add_lora("tuned", rank=4, seed=99)
            outs = engine.generate(
                [PROMPT], SamplingParams(max_tokens=4, temperature=0.0)
            )
            assert outs[0].output_token_ids == base1.output_token_ids
            req2 = engine.add_request(
                PROMPT, SamplingParams(max_tokens=4, temperature=0.0),
                lora_name="tuned",
            )
            out2 = None
            while engine.has_unfinished():
                for o in engine.step():
                    if o.finished and o.request_id == req2:
                        out2 = o
            # TP=2 adapter run must match the TP=1 adapter run (sharded
            # adapter reconstructions agree), and differ from base
            assert out2.output_token_ids == lora1.output_token_ids
            assert out2.output_token_ids != base1.output_token_ids
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_tp2_lora_matches_tp1():
    """LoRA under TP=2: the driver's add_lora broadcast registers the shard
    on the worker; generation with the adapter matches TP=1 exactly."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29619
        procs = [
            ctx.Process(target=_tp_lora_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# ------------------------------------------------------ PP (pipeline stages)
def _pp_engine_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams

    # single-process reference first (per-module seeded init makes the
    # weights identical regardless of the PP layout)
    ps.ensure_single_process()
    torch.manual_seed(321)
    eng1 = LLMEngine(_engine_cfg(), device="cpu")
    ref_out = eng1.generate(
        [PROMPT, [9, 9, 2] * 8],
        SamplingParams(max_tokens=4, temperature=0.0),
    )
    ref_tokens = [o.output_token_ids for o in ref_out]
    ps.destroy()

    _init(rank, 2, port)
    try:
        torch.manual_seed(321)
        cfg = _engine_cfg()
        cfg.parallel.pipeline_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        # stage 0 holds the embedding + first half, stage 1 the rest
        assert engine.runner.model.num_local_layers * 2 == \
            cfg.model.num_layers
        if engine.is_driver:
            outs = engine.generate(
                [PROMPT, [9, 9, 2] * 8],
                SamplingParams(max_tokens=4, temperature=0.0),
            )
            assert [o.output_token_ids for o in outs] == ref_tokens
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_pp2_engine_matches_single_process():
    """Pipeline parallelism (2 stages over gloo): stage-sliced layers,
    activations via p2p send/recv, logits shipped back to the driver —
    token-exact vs the single-process engine."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29621
        procs = [
            ctx.Process(target=_pp_engine_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# ------------------------------------------------- EP (experts over TP group)
def _ep_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.block_manager import BlockManager
    from fusioninfer_amd.engine.model_runner import ModelRunner
    from fusioninfer_amd.engine.sequence import SamplingParams, Sequence
    from fusioninfer_amd.models.registry import get_model_config

    def moe_cfg():
        cfg = _engine_cfg()
        cfg.model = get_model_config("tiny-qwen3-moe")
        return cfg

    ps.ensure_single_process()
    torch.manual_seed(77)
    cfg1 = moe_cfg()
    runner1 = ModelRunner(cfg1, "cpu")
    runner1.allocate_kv_caches()
    bm1 = BlockManager(64, 16)
    seq = Sequence("s", PROMPT, SamplingParams())
    bm1.allocate(seq)
    ref_logits = runner1.execute_prefill([seq], bm1).float()
    ps.destroy()

    _init(rank, 2, port)
    try:
        ps.init_distributed(tensor_parallel_size=2, backend="gloo")
        torch.manual_seed(77)
        cfg2 = moe_cfg()
        cfg2.parallel.tensor_parallel_size = 2
        runner2 = ModelRunner(cfg2, "cpu")
        # experts are sharded across the group (EP): 4 of 8 per rank
        assert runner2.model.layers[0].mlp.gate_up_t.shape[0] == 4
        runner2.allocate_kv_caches()
        bm2 = BlockManager(64, 16)
        seq2 = Sequence("s2", PROMPT, SamplingParams())
        bm2.allocate(seq2)
        tp_logits = runner2.execute_prefill([seq2], bm2).float()
        rel = (tp_logits - ref_logits).norm() / ref_logits.norm()
        assert rel.item() < 0.05, rel.item()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_ep2_moe_matches_single_rank():
    """MoE experts sharded over a 2-rank group (allreduce-combine EP):
    logits match the single-rank model (same per-expert seeds)."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29623
        procs = [
            ctx.Process(target=_ep_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# --------------------------------------- speculative decoding under TP=2
def _tp_spec_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.engine.spec_decode import SpeculativeConfig

    spec_prompt = [3, 1, 4, 1, 5, 9] * 5  # repetition so drafts fire

    # TP=1 greedy reference
    ps.ensure_single_process()
    torch.manual_seed(5)
    eng1 = LLMEngine(_engine_cfg(), device="cpu")
    ref = eng1.generate([spec_prompt],
                        SamplingParams(max_tokens=12, temperature=0.0))[0]
    ps.destroy()

    _init(rank, 2, port)
    try:
        torch.manual_seed(5)
        cfg = _engine_cfg()
        cfg.parallel.tensor_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        cfg.speculative = SpeculativeConfig(num_speculative_tokens=4)
        engine = LLMEngine(cfg, device="cpu")
        if engine.is_driver:
            outs = engine.generate(
                [spec_prompt], SamplingParams(max_tokens=12, temperature=0.0)
            )
            # the broadcast verify payload ran on both ranks (per-layer
            # all-reduces lined up) and acceptance is token-exact
            assert outs[0].output_token_ids == ref.output_token_ids
            assert engine.num_spec_draft_tokens > 0
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_tp2_spec_decode_matches_tp1():
    """Speculative verify payloads are plain prefill payloads: the TP
    driver broadcasts them and workers execute the same forward."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29631
        procs = [
            ctx.Process(target=_tp_spec_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# ------------------------------------ PP microbatch-overlapped prefill
def _pp_microbatch_worker(rank, port, results):
    from fusioninfer_amd.config import CacheConfig, SchedulerConfig
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams

    def big_cfg():
        return EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=256),
            scheduler=SchedulerConfig(
                max_num_seqs=8, max_num_batched_tokens=2048,
                max_model_len=512,
            ),
            seed=11,
        )

    # 4 prompts x ~150 tokens = one ~600-token admission step, which the
    # splitter cuts into 2 microbatches (>= 256 tokens each)
    prompts = [
        [(7 * i + j) % 97 + 2 for j in range(150)] for i in range(4)
    ]

    ps.ensure_single_process()
    torch.manual_seed(77)
    eng1 = LLMEngine(big_cfg(), device="cpu")
    ref = [
        o.output_token_ids
        for o in eng1.generate(prompts,
                               SamplingParams(max_tokens=4, temperature=0.0))
    ]
    ps.destroy()

    _init(rank, 2, port)
    try:
        torch.manual_seed(77)
        cfg = big_cfg()
        cfg.parallel.pipeline_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        # the admission step really is split (assert on the driver side
        # by probing the splitter with the same batch shape)
        if engine.is_driver:
            subs = engine.runner._split_prefill_payload({
                "kind": "prefill",
                "ids": [0] * 600, "positions": [0] * 600,
                "slots": [0] * 600, "cu": [0, 150, 300, 450, 600],
                "new_lens": [150] * 4, "total_lens": [150] * 4,
                "bt": [[0]] * 4, "sample": [True] * 4,
                "lora_names": [None] * 4,
            })
            assert len(subs) == 2
            assert [len(s["ids"]) for s in subs] == [300, 300]
            outs = engine.generate(
                prompts, SamplingParams(max_tokens=4, temperature=0.0)
            )
            assert [o.output_token_ids for o in outs] == ref
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_pp2_microbatch_prefill_matches_single_process():
    """Microbatch-overlapped PP prefill (pre-posted logits irecv, isend
    activations): token-exact vs the single-process engine."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29637
        procs = [
            ctx.Process(target=_pp_microbatch_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# ----------------------------------- one-shot all-reduce (small tensors)
def _oneshot_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps

    _init(rank, 2, port)
    try:
        ps.init_distributed(tensor_parallel_size=2, backend="gloo")
        # small fp32 tensor -> one-shot (all-gather + local sum) path
        small = torch.full((64,), float(rank + 1))
        out = ps.tp_all_reduce(small.clone())
        assert torch.equal(out, torch.full((64,), 3.0))
        # large fp32 tensor -> ring all-reduce path; same result contract
        big = torch.full((200_000,), float(rank + 1))
        out = ps.tp_all_reduce(big.clone())
        assert torch.equal(out, torch.full((200_000,), 3.0))
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_tp2_oneshot_allreduce_matches_ring():
    """Small latency-bound tensors take the one-shot (all-gather + sum)
    path — 1 p2p step on near-fully-connected xGMI vs the ring's
    2*(N-1); both paths must produce the exact sum."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29641
        procs = [
            ctx.Process(target=_oneshot_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# --------------------------------------- speculative decoding under PP=2
def _pp_spec_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.engine.spec_decode import SpeculativeConfig

    prompts = [[3, 1, 4, 1, 5, 9] * 5, [2, 7, 2, 7] * 8]

    ps.ensure_single_process()
    torch.manual_seed(13)
    eng1 = LLMEngine(_engine_cfg(), device="cpu")
    ref = [o.output_token_ids
           for o in eng1.generate(prompts,
                                  SamplingParams(max_tokens=12,
                                                 temperature=0.0))]
    ps.destroy()

    _init(rank, 2, port)
    try:
        torch.manual_seed(13)
        cfg = _engine_cfg()
        cfg.parallel.pipeline_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        cfg.speculative = SpeculativeConfig(num_speculative_tokens=4)
        engine = LLMEngine(cfg, device="cpu")
        if engine.is_driver:
            outs = engine.generate(
                prompts, SamplingParams(max_tokens=12, temperature=0.0)
            )
            assert [o.output_token_ids for o in outs] == ref
            assert engine.num_spec_draft_tokens > 0
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_pp2_spec_decode_matches_single_process():
    """Spec verify payloads flow through the PP stages (and, when large
    enough, the microbatch splitter rebases logits_rows): token-exact."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29643
        procs = [
            ctx.Process(target=_pp_spec_worker, args=(r, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert results[0] == "ok" and results[1] == "ok"


# ------------------------------------------------ TP x PP composition
def _tp_pp_worker(rank, port, results):
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams

    prompts = [PROMPT, [9, 9, 2] * 8]

    ps.ensure_single_process()
    torch.manual_seed(31)
    eng1 = LLMEngine(_engine_cfg(), device="cpu")
    ref = [o.output_token_ids
           for o in eng1.generate(prompts,
                                  SamplingParams(max_tokens=6,
                                                 temperature=0.0))]
    ps.destroy()

    _init(rank, 4, port)
    try:
        torch.manual_seed(31)
        cfg = _engine_cfg()
        cfg.parallel.tensor_parallel_size = 2
        cfg.parallel.pipeline_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        # stage = rank // tp; each stage holds half the layers, each tp
        # rank half the heads
        assert engine.runner.model.num_local_layers * 2 == \
            cfg.model.num_layers
        if engine.is_driver:
            outs = engine.generate(
                prompts, SamplingParams(max_tokens=6, temperature=0.0)
            )
            assert [o.output_token_ids for o in outs] == ref
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_tp2_pp2_engine_matches_single_process():
    """TP x PP (world 4, 2 stages x 2 shards over gloo): stage-contiguous
    TP groups, column-wise activation p2p — token-exact vs 1 process."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29647
        procs = [
            ctx.Process(target=_tp_pp_worker, args=(r, port, results))
            for r in range(4)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
            assert p.exitcode == 0, f"rank exited {p.exitcode}"
        assert all(results[r] == "ok" for r in range(4))


# ------------------------------------------------ PD x TP (world=4, gloo)
def _pd_tp_worker(rank, port, results):
    """TP=2 prefiller (ranks 0,1) -> TP=2 decoder (ranks 2,3): each rank
    ships/receives its OWN KV shard over a dedicated pair group (0<->2,
    1<->3), never the default group (VERDICT round-1 item 8). Token-exact
    vs a monolithic TP=1 engine on the same seeded weights."""
    import types

    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.pd import build_pd_connector

    # monolithic TP=1 reference first (no process group yet)
    expected = None
    if rank == 2:
        ps.ensure_single_process()
        mono = LLMEngine(_engine_cfg(), device="cpu")
        outs = mono.generate(
            [PROMPT], SamplingParams(max_tokens=N_TOKENS, temperature=0.0)
        )
        expected = outs[0].output_token_ids
        ps.destroy()

    _init(rank, 4, port)
    try:
        ps.init_distributed(tensor_parallel_size=2, backend="gloo")
        cfg = _engine_cfg()
        cfg.parallel.tensor_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        kvt = types.SimpleNamespace(
            kv_connector="RcclConnector",
            kv_role="kv_producer" if rank < 2 else "kv_consumer",
            kv_rank=rank // 2, kv_world_size=2,
        )
        conn = build_pd_connector(kvt, device="cpu", tp=2)
        engine.kv_connector = conn

        if rank in (1, 3):       # TP workers: driven by broadcasts
            engine.worker_loop()
            results[rank] = "ok"
            return
        if rank == 0:            # prefiller driver
            req_id = engine.add_export_request(PROMPT)
            while engine.has_unfinished():
                engine.step()
            seq = engine._held[req_id]
            first = seq.output_token_ids[0]
            engine.pd_send_held(conn, req_id, len(PROMPT), first, tag=7)
            engine.release_held(req_id)
            engine.stop_workers()
            results[0] = "ok"
            return
        # rank == 2: decoder driver
        def alloc(n):
            ids = engine.allocate_import_blocks(n)
            engine.pd_recv_broadcast(ids)
            return ids

        _, prompt_len, first, tag = conn.recv_kv(
            engine.runner.kv_caches, alloc
        )
        assert tag == 7 and prompt_len == len(PROMPT)
        req_id = engine.add_imported_request(
            prompt_len, first,
            SamplingParams(max_tokens=N_TOKENS, temperature=0.0),
        )
        got = {}
        while engine.has_unfinished():
            for out in engine.step():
                if out.finished:
                    got[out.request_id] = out
        toks = got[req_id].output_token_ids
        assert toks == expected, (toks, expected)
        engine.stop_workers()
        results[2] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_pd_tp2_shard_handoff_token_exact():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29617
        procs = [
            ctx.Process(target=_pd_tp_worker, args=(r, port, results))
            for r in range(4)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0, f"a rank exited {p.exitcode}"
        assert all(results.get(r) == "ok" for r in range(4)), dict(results)


# --------------------------------------------------- TP=8 launch (world=8)
def _tp8_worker(rank, port, results):
    """TP=8 engine end-to-end on gloo (BASELINE config #5 launch shape:
    Llama-70B-class TP=8 — exercised on a small config so the 8-way
    sharding/collective paths are launch-robust before the driver's
    8-GPU runs; VERDICT round-1 item 2)."""
    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams

    _init(rank, 8, port)
    try:
        from fusioninfer_amd.config import (
            CacheConfig, EngineConfig, SchedulerConfig,
        )
        mc = get_model_config("Qwen3-0.6B")  # 16 heads / 8 kv: TP=8 divides
        mc.num_layers = 2
        cfg = EngineConfig(
            model=mc,
            cache=CacheConfig(num_gpu_blocks=32),
            scheduler=SchedulerConfig(
                max_num_seqs=2, max_num_batched_tokens=256, max_model_len=96
            ),
        )
        cfg.parallel.tensor_parallel_size = 8
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        if engine.is_driver:
            outs = engine.generate(
                [PROMPT], SamplingParams(max_tokens=3, temperature=0.0)
            )
            assert len(outs[0].output_token_ids) == 3
            engine.stop_workers()
        else:
            engine.worker_loop()
        results[rank] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_tp8_engine_generates():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29619
        procs = [
            ctx.Process(target=_tp8_worker, args=(r, port, results))
            for r in range(8)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0, f"a rank exited {p.exitcode}"
        assert all(results.get(r) == "ok" for r in range(8)), dict(results)


# ------------------------------------------------ PD x PP (world=4, gloo)
def _pd_pp_worker(rank, port, results):
    """PP=2 prefiller (stages 0,1 = ranks 0,1) -> PP=2 decoder (ranks
    2,3): each STAGE ships its own layers' KV shard over its pair group
    (0<->2, 1<->3) — the PD x PP composition round 1 left unvalidated."""
    import types

    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.pd import build_pd_connector

    expected = None
    if rank == 2:
        ps.ensure_single_process()
        mono = LLMEngine(_engine_cfg(), device="cpu")
        outs = mono.generate(
            [PROMPT], SamplingParams(max_tokens=N_TOKENS, temperature=0.0)
        )
        expected = outs[0].output_token_ids
        ps.destroy()

    _init(rank, 4, port)
    try:
        ps.init_distributed(tensor_parallel_size=1, backend="gloo",
                            pipeline_parallel_size=2)
        cfg = _engine_cfg()
        cfg.parallel.pipeline_parallel_size = 2
        cfg.parallel.distributed_backend = "gloo"
        engine = LLMEngine(cfg, device="cpu")
        kvt = types.SimpleNamespace(
            kv_connector="RcclConnector",
            kv_role="kv_producer" if rank < 2 else "kv_consumer",
            kv_rank=rank // 2, kv_world_size=2,
        )
        # exec world per side = pp*tp = 2: pair groups (0,2) and (1,3)
        conn = build_pd_connector(kvt, device="cpu", tp=2)
        engine.kv_connector = conn

        if rank in (1, 3):   # stage-1 workers
            engine.worker_loop()
            results[rank] = "ok"
            return
        if rank == 0:        # prefiller driver (stage 0)
            req_id = engine.add_export_request(PROMPT)
            while engine.has_unfinished():
                engine.step()
            seq = engine._held[req_id]
            first = seq.output_token_ids[0]
            engine.pd_send_held(conn, req_id, len(PROMPT), first, tag=9)
            engine.release_held(req_id)
            engine.stop_workers()
            results[0] = "ok"
            return

        def alloc(n):
            ids = engine.allocate_import_blocks(n)
            engine.pd_recv_broadcast(ids)
            return ids

        _, prompt_len, first, tag = conn.recv_kv(
            engine.runner.kv_caches, alloc
        )
        assert tag == 9 and prompt_len == len(PROMPT)
        req_id = engine.add_imported_request(
            prompt_len, first,
            SamplingParams(max_tokens=N_TOKENS, temperature=0.0),
        )
        got = {}
        while engine.has_unfinished():
            for out in engine.step():
                if out.finished:
                    got[out.request_id] = out
        toks = got[req_id].output_token_ids
        assert toks == expected, (toks, expected)
        engine.stop_workers()
        results[2] = "ok"
    finally:
        dist.destroy_process_group()
        ps.destroy()


def test_pd_pp2_stage_shard_handoff_token_exact():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29621
        procs = [
            ctx.Process(target=_pd_pp_worker, args=(r, port, results))
            for r in range(4)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0, f"a rank exited {p.exitcode}"
        assert all(results.get(r) == "ok" for r in range(4)), dict(results)
