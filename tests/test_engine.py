"""CPU tests of the generation engine (paged KV + continuous batching)
vs a plain full-attention reference forward."""

import pytest
import torch

from distllm_amd.engine import LLMEngine, SamplingParams
from distllm_amd.models import build_model


@pytest.fixture(scope="module")
def model():
    torch.manual_seed(7)
    return build_model("llama-tiny", dtype=torch.float32)


def _ref_greedy(model, prompt_ids, n_new):
    """Greedy decode via repeated full forward (no KV cache)."""
    ids = list(prompt_ids)
    out = []
    for _ in range(n_new):
        t = torch.tensor([ids])
        mask = torch.ones_like(t, dtype=torch.int32)
        hidden = model(t, mask)
        logits = model.logits(hidden[:, -1])
        nxt = int(logits.argmax(-1))
        out.append(nxt)
        ids.append(nxt)
    return out


def test_engine_matches_full_forward_greedy(model):
    engine = LLMEngine(model, max_batch_size=4, max_model_len=128,
                       num_blocks=64)
    params = SamplingParams(temperature=0.0, min_p=0.0, max_tokens=8,
                            stop_token_ids=())
    prompts = [[5, 9, 100, 42], [7, 7, 3, 900, 12, 4]]
    outs = engine.generate(prompts, params)
    for p, o in zip(prompts, outs):
        ref = _ref_greedy(model, p, 8)
        assert o == ref, f"engine {o} != ref {ref}"


def test_engine_continuous_batching_many(model):
    engine = LLMEngine(model, max_batch_size=3, max_model_len=64,
                       num_blocks=48)
    params = SamplingParams(temperature=0.0, min_p=0.0, max_tokens=4)
    prompts = [[i + 5, i + 6, i + 7] for i in range(7)]
    outs = engine.generate(prompts, params)
    assert len(outs) == 7
    for p, o in zip(prompts, outs):
        assert o == _ref_greedy(model, p, 4)


def test_engine_blocks_released(model):
    engine = LLMEngine(model, max_batch_size=2, max_model_len=64,
                       num_blocks=32)
    free0 = len(engine.blockman.free)
    params = SamplingParams(temperature=0.0, max_tokens=3, min_p=0.0)
    engine.generate([[1, 2, 3]], params)
    assert len(engine.blockman.free) == free0


def test_native_generator_end_to_end():
    from distllm_amd.generate import get_generator

    gen = get_generator({
        "name": "native", "model": "llama-tiny", "temperature": 0.0,
        "min_p": 0.0, "max_tokens": 5, "max_model_len": 128,
        "max_batch_size": 4,
    })
    out = gen.generate(["Hello world test", "Another prompt here longer"])
    assert len(out) == 2
    assert all(isinstance(x, str) for x in out)


def test_vllm_alias_maps_to_native():
    from distllm_amd.generate.generators import STRATEGIES

    assert STRATEGIES["vllm"][1].__name__ == "NativeGenerator"


def test_dynamic_batcher_groups_requests():
    from concurrent.futures import ThreadPoolExecutor

    from distllm_amd.engine.batcher import DynamicBatcher

    calls = []

    class FakeGen:
        def generate(self, prompts):
            calls.append(len(prompts))
            return [p.upper() for p in prompts]

    b = DynamicBatcher(FakeGen(), max_batch=8, window_ms=50)
    with ThreadPoolExecutor(max_workers=6) as ex:
        futs = [ex.submit(b.generate_one, f"p{i}") for i in range(6)]
        outs = [f.result(timeout=30) for f in futs]
    assert sorted(outs) == [f"P{i}" for i in range(6)]
    # concurrent requests were grouped (fewer calls than requests)
    assert sum(calls) == 6 and len(calls) < 6
    b.close()


def test_dynamic_batcher_error_propagation():
    from distllm_amd.engine.batcher import DynamicBatcher

    class BadGen:
        def generate(self, prompts):
            raise RuntimeError("engine down")

    b = DynamicBatcher(BadGen(), window_ms=1)
    import pytest as _pytest

    with _pytest.raises(RuntimeError):
        b.generate_one("x", timeout=10)
    b.close()


def test_engine_truncates_long_prompts():
    import torch

    from distllm_amd.engine import LLMEngine, SamplingParams
    from distllm_amd.models import build_model

    torch.manual_seed(3)
    m = build_model("llama-tiny", dtype=torch.float32)
    engine = LLMEngine(m, max_batch_size=2, max_model_len=32,
                       num_blocks=16)
    params = SamplingParams(temperature=0.0, min_p=0.0, max_tokens=2)
    long_prompt = list(range(5, 105))  # 100 tokens >> max_model_len
    outs = engine.generate([long_prompt], params)
    assert len(outs[0]) >= 1  # generated within the truncated window


def test_engine_loop_online_batching():
    """Staggered concurrent requests through the online loop match
    single-shot greedy generation exactly."""
    import time as _time
    from concurrent.futures import ThreadPoolExecutor

    from distllm_amd.engine import LLMEngine, SamplingParams
    from distllm_amd.engine.loop import EngineLoop
    from distllm_amd.models import build_model

    torch.manual_seed(7)
    m = build_model("llama-tiny", dtype=torch.float32)
    params = SamplingParams(temperature=0.0, min_p=0.0, max_tokens=64)
    ref_engine = LLMEngine(m, max_batch_size=4, max_model_len=128,
                           num_blocks=64)
    prompts = [[5, 9, 100], [7, 3, 900, 12], [42, 42], [1, 2, 3, 4, 5]]
    want = [ref_engine.generate([p],
                                SamplingParams(temperature=0.0, min_p=0.0,
                                               max_tokens=6))[0]
            for p in prompts]

    engine = LLMEngine(m, max_batch_size=3, max_model_len=128,
                       num_blocks=64)
    loop = EngineLoop(engine, params)
    try:
        with ThreadPoolExecutor(max_workers=4) as ex:
            futs = []
            for i, p in enumerate(prompts):
                futs.append(ex.submit(loop.generate, [p], 6))
                _time.sleep(0.01)  # stagger arrivals mid-generation
            outs = [f.result(timeout=60)[0] for f in futs]
        assert outs == want, f"{outs} != {want}"
    finally:
        loop.close()


def test_native_generator_loop_path():
    from distllm_amd.generate import get_generator

    gen = get_generator({"name": "native", "model": "llama-tiny",
                         "temperature": 0.0, "min_p": 0.0,
                         "max_tokens": 4, "max_model_len": 128,
                         "max_batch_size": 4})
    direct = gen.generate(["hello world"])
    gen.get_loop()
    via_loop = gen.generate(["hello world"])  # routes through the loop
    one = gen.generate_one("hello world")
    assert direct == via_loop and one == direct[0]


def test_beam_search():
    """Beam search: beam=1 equals greedy argmax; a wider beam never
    returns a lower length-normalized logprob than greedy."""
    import torch

    from distllm_amd.engine.beam import beam_generate
    from distllm_amd.models import build_model

    torch.manual_seed(11)
    m = build_model("llama-tiny", device="cpu", dtype=torch.float32)
    prompt = [5, 9, 100, 42]

    def seq_logprob(ids):
        lp = 0.0
        cur = list(prompt)
        for t in ids:
            x = torch.tensor([cur])
            mask = torch.ones_like(x, dtype=torch.int32)
            logits = m.logits(m(x, mask)[:, -1]).float()
            lp += float(torch.log_softmax(logits, -1)[0, t])
            cur.append(t)
        return lp / max(1, len(ids))

    b1 = beam_generate(m, prompt, num_beams=1, max_tokens=5)
    # greedy reference
    greedy = []
    cur = list(prompt)
    for _ in range(5):
        x = torch.tensor([cur])
        mask = torch.ones_like(x, dtype=torch.int32)
        t = int(m.logits(m(x, mask)[:, -1]).argmax(-1))
        greedy.append(t)
        cur.append(t)
    assert b1 == greedy

    b4 = beam_generate(m, prompt, num_beams=4, max_tokens=5)
    assert len(b4) == 5
    assert seq_logprob(b4) >= seq_logprob(greedy) - 1e-6


def test_native_generator_beam_path():
    from distllm_amd.generate import get_generator

    gen = get_generator({"name": "huggingface", "model": "llama-tiny",
                         "num_beams": 3, "max_tokens": 4,
                         "max_batch_size": 2, "max_model_len": 128})
    out = gen.generate(["hello world"])
    assert len(out) == 1 and isinstance(out[0], str)


def test_block_manager_invariants():
    """Allocator invariants: block 0 reserved, exhaustion raises,
    release returns capacity, blocks_for rounds up."""
    from distllm_amd.engine.engine import BlockManager

    bm = BlockManager(num_blocks=8, blk=16)
    assert 0 not in bm.free  # dummy graph-slot block stays reserved
    assert bm.blocks_for(1) == 1 and bm.blocks_for(16) == 1
    assert bm.blocks_for(17) == 2 and bm.blocks_for(0) == 0
    assert bm.can_alloc(7) and not bm.can_alloc(8)
    a = bm.alloc(3)
    b = bm.alloc(4)
    assert len(set(a) | set(b)) == 7  # disjoint
    import pytest as _p
    with _p.raises(RuntimeError):
        bm.alloc(1)
    bm.release(a)
    assert bm.can_alloc(3) and not bm.can_alloc(4)
