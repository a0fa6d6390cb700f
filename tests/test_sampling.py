"""Sampling feature parity: penalties, logit_bias, min_p, per-request seed."""
import torch

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.model_runner import Sampler
from gpustack_amd.engine.sequence import Sequence


def _eng():
    return LLMEngine(EngineConfig(model="tiny", device="cpu",
                                  kv_cache_blocks=64, max_model_len=128))


def test_logit_bias_forces_token():
    eng = _eng()
    p = SamplingParams(max_tokens=4, ignore_eos=True,
                       logit_bias={7: 1000.0})
    out = eng.generate([[1, 2, 3]], p)[0]
    assert out == [7, 7, 7, 7]


def test_presence_penalty_breaks_repetition():
    eng = _eng()
    plain = eng.generate([[1, 2, 3]],
                         SamplingParams(max_tokens=12, ignore_eos=True))[0]
    eng2 = _eng()
    pen = eng2.generate([[1, 2, 3]],
                        SamplingParams(max_tokens=12, ignore_eos=True,
                                       presence_penalty=50.0))[0]
    # a -50 presence penalty makes every emitted token unrepeatable
    assert len(set(pen)) == 12
    assert len(set(plain)) <= 12


def test_repetition_penalty_processor():
    s = Sequence("t", [1], SamplingParams(repetition_penalty=2.0))
    s.output_token_ids = [3]
    row = torch.tensor([0.0, 1.0, -1.0, 4.0])
    out = Sampler('cpu')._process_logits(row.clone(), s)
    assert out[3] == 2.0 and out[1] == 1.0
    s2 = Sequence("t", [1], SamplingParams(frequency_penalty=0.5))
    s2.output_token_ids = [2, 2, 2]
    out2 = Sampler('cpu')._process_logits(row.clone(), s2)
    assert out2[2] == -1.0 - 1.5


def test_seeded_sampling_deterministic():
    p = SamplingParams(max_tokens=8, ignore_eos=True, temperature=0.8, seed=42)
    a = _eng().generate([[1, 2, 3]], p)[0]
    b = _eng().generate([[1, 2, 3]], p)[0]
    assert a == b
    c = _eng().generate([[1, 2, 3]],
                        SamplingParams(max_tokens=8, ignore_eos=True,
                                       temperature=0.8, seed=43))[0]
    assert a != c  # different seed diverges (overwhelmingly likely)


def test_min_p_restricts_support():
    p = SamplingParams(max_tokens=6, ignore_eos=True, temperature=5.0,
                       min_p=1.0, seed=0)
    greedy = _eng().generate([[1, 2, 3]],
                             SamplingParams(max_tokens=6, ignore_eos=True))[0]
    # min_p = 1.0 collapses high-temperature sampling onto the argmax token
    out = _eng().generate([[1, 2, 3]], p)[0]
    assert out == greedy


def test_guided_choice_exact():
    eng = _eng()
    choices = ((41, 42, 43), (44, 45))
    p = SamplingParams(max_tokens=10, guided_token_seqs=choices,
                       eos_token_id=1)
    out = eng.generate([[1, 2, 3]], p)[0]
    body = out[:-1] if out and out[-1] == 1 else out
    assert tuple(body) in choices


def test_guided_choice_single_forces_sequence():
    eng = _eng()
    p = SamplingParams(max_tokens=10, guided_token_seqs=((9, 8, 7, 6),),
                       eos_token_id=1)
    out = eng.generate([[5]], p)[0]
    assert out[:4] == [9, 8, 7, 6]


def test_top_logprobs():
    """OpenAI top_logprobs: per-token top-k alternatives; the chosen
    (greedy) token is the top-1 alternative with a matching logprob."""
    import math

    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64))
    p = SamplingParams(max_tokens=4, ignore_eos=True, logprobs=True,
                       top_logprobs=3)
    rid = eng.add_request([1, 2, 3], p)
    outs = []
    while eng.has_unfinished():
        outs.extend(o for o in eng.step() if o.request_id == rid)
    assert len(outs) == 4
    for o in outs:
        assert o.logprob is not None
        assert o.top_logprobs is not None and len(o.top_logprobs) == 3
        ids = [t for t, _ in o.top_logprobs]
        lps = [lp for _, lp in o.top_logprobs]
        assert ids[0] == o.token_id          # greedy pick is argmax
        assert math.isclose(lps[0], o.logprob, rel_tol=1e-5, abs_tol=1e-6)
        assert lps == sorted(lps, reverse=True)
        assert all(lp <= 0 for lp in lps)


def test_min_tokens_suppresses_eos():
    """min_tokens (vLLM semantics): EOS/stop finishes are ignored until the
    sequence has produced that many tokens."""
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams

    eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64))
    # force EOS every step via logit_bias toward the eos id
    eos = eng.cfg.spec.eos_token_id
    p = SamplingParams(max_tokens=10, logit_bias={eos: 100.0})
    out = eng.generate([[2, 3, 4]], p)[0]
    assert len(out) == 1 and out[0] == eos  # stops immediately without min
    p = SamplingParams(max_tokens=10, min_tokens=5, logit_bias={eos: 100.0})
    out = eng.generate([[2, 3, 4]], p)[0]
    assert len(out) == 6  # 5 masked-EOS steps + the terminating EOS
    # EOS is masked out of the distribution (not merely ignored by the
    # finish check): no EOS appears in user-visible output before the end
    assert eos not in out[:5]
    assert out[5] == eos
