"""Cross-encoder reranker family (models/encoder.py): logits-exact vs HF
transformers at fp32 on CPU for both BERT and XLM-Roberta heads, plus the
pair-scoring runner semantics (padding invariance, truncation, ordering)."""
import torch

from gpustack_amd.models.encoder import (
    CrossEncoderModel, CrossEncoderRunner, EncoderSpec, is_encoder_arch,
)

BERT_SPEC = EncoderSpec(
    architecture="BertForSequenceClassification", vocab_size=256,
    hidden_size=64, intermediate_size=128, num_layers=3, num_heads=4,
    max_position_embeddings=96, type_vocab_size=2, num_labels=1,
    pad_token_id=0,
)
ROBERTA_SPEC = EncoderSpec(
    architecture="XLMRobertaForSequenceClassification", vocab_size=256,
    hidden_size=64, intermediate_size=128, num_layers=3, num_heads=4,
    max_position_embeddings=96, type_vocab_size=1, num_labels=1,
    pad_token_id=1,
)


def test_is_encoder_arch():
    assert is_encoder_arch("BertForSequenceClassification")
    assert is_encoder_arch("XLMRobertaForSequenceClassification")
    assert not is_encoder_arch("LlamaForCausalLM")


def _hf_bert(spec: EncoderSpec):
    from transformers import BertConfig, BertForSequenceClassification

    cfg = BertConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        intermediate_size=spec.intermediate_size,
        max_position_embeddings=spec.max_position_embeddings,
        type_vocab_size=spec.type_vocab_size,
        layer_norm_eps=spec.layer_norm_eps, num_labels=spec.num_labels,
        pad_token_id=spec.pad_token_id, attn_implementation="eager",
    )
    return BertForSequenceClassification(cfg).eval().float()


def _hf_roberta(spec: EncoderSpec):
    from transformers import (XLMRobertaConfig,
                              XLMRobertaForSequenceClassification)

    cfg = XLMRobertaConfig(
        vocab_size=spec.vocab_size, hidden_size=spec.hidden_size,
        num_hidden_layers=spec.num_layers,
        num_attention_heads=spec.num_heads,
        intermediate_size=spec.intermediate_size,
        max_position_embeddings=spec.max_position_embeddings,
        type_vocab_size=spec.type_vocab_size,
        layer_norm_eps=spec.layer_norm_eps, num_labels=spec.num_labels,
        pad_token_id=spec.pad_token_id, attn_implementation="eager",
    )
    return XLMRobertaForSequenceClassification(cfg).eval().float()


def test_bert_matches_hf_logits():
    torch.manual_seed(0)
    hf = _hf_bert(BERT_SPEC)
    m = CrossEncoderModel(BERT_SPEC)
    m.load_hf_state_dict(hf.state_dict())

    ids = torch.tensor([[101, 7, 8, 9, 102, 30, 31, 102, 0, 0],
                        [101, 5, 102, 40, 102, 0, 0, 0, 0, 0]])
    mask = (ids != 0).long()
    types = torch.tensor([[0, 0, 0, 0, 0, 1, 1, 1, 0, 0],
                          [0, 0, 0, 1, 1, 0, 0, 0, 0, 0]])
    with torch.inference_mode():
        want = hf(input_ids=ids, attention_mask=mask,
                  token_type_ids=types).logits
    got = m(ids, mask, types)
    assert torch.allclose(got, want, atol=2e-5, rtol=1e-4), \
        (got - want).abs().max()


def test_xlm_roberta_matches_hf_logits():
    torch.manual_seed(1)
    hf = _hf_roberta(ROBERTA_SPEC)
    m = CrossEncoderModel(ROBERTA_SPEC)
    m.load_hf_state_dict(hf.state_dict())

    ids = torch.tensor([[0, 7, 8, 9, 2, 2, 30, 31, 2, 1],
                        [0, 5, 2, 2, 40, 2, 1, 1, 1, 1]])
    mask = (ids != 1).long()
    with torch.inference_mode():
        want = hf(input_ids=ids, attention_mask=mask).logits
    got = m(ids, mask)
    assert torch.allclose(got, want, atol=2e-5, rtol=1e-4), \
        (got - want).abs().max()


def test_runner_scores_and_padding_invariance():
    """Batch padding must not change a pair's score, and the runner's join
    layout must keep scores deterministic under document reordering."""
    torch.manual_seed(2)
    m = CrossEncoderModel(BERT_SPEC)
    m.random_init(seed=3)
    r = CrossEncoderRunner(m, cls_id=101, sep_id=102)
    q = [7, 8, 9]
    docs = [[30, 31, 32, 33], [40], [50, 51]]
    scores = r.score(q, docs)
    assert len(scores) == 3
    # one doc alone == same doc in a batch (padding invariance)
    alone = r.score(q, [docs[1]])[0]
    assert abs(alone - scores[1]) < 1e-5
    # reordering docs permutes scores identically
    perm = r.score(q, [docs[2], docs[0], docs[1]])
    assert abs(perm[0] - scores[2]) < 1e-5
    assert abs(perm[1] - scores[0]) < 1e-5


def test_runner_truncates_long_pairs():
    m = CrossEncoderModel(BERT_SPEC)
    m.random_init(seed=4)
    r = CrossEncoderRunner(m, cls_id=101, sep_id=102)
    q = list(range(3, 60))
    d = list(range(60, 200))
    [s] = r.score(q, [d])  # would exceed max_position_embeddings untruncated
    assert isinstance(s, float)


def test_spec_from_hf_config():
    spec = EncoderSpec.from_hf_config({
        "architectures": ["XLMRobertaForSequenceClassification"],
        "vocab_size": 250002, "hidden_size": 1024,
        "num_hidden_layers": 24, "num_attention_heads": 16,
        "intermediate_size": 4096, "max_position_embeddings": 8194,
        "type_vocab_size": 1, "layer_norm_eps": 1e-05,
        "id2label": {"0": "LABEL_0"}, "pad_token_id": 1,
    })  # bge-reranker-v2-m3's config shape
    assert spec.is_roberta and spec.num_labels == 1
    assert spec.pad_token_id == 1 and spec.hidden_size == 1024


def test_encoder_serving_mode(tmp_path):
    """engine_server detects a sequence-classification checkpoint and
    serves the cross-encoder /v1/rerank + /v1/score instead of an LLM
    engine; scores over the wire match the in-process runner exactly."""
    import json
    import math
    import socket
    import subprocess
    import sys
    import time

    import httpx
    from safetensors.torch import save_file

    # a real (seeded) tiny BERT checkpoint on disk
    torch.manual_seed(7)
    m = CrossEncoderModel(BERT_SPEC)
    m.random_init(seed=7)
    cfgd = {
        "architectures": ["BertForSequenceClassification"],
        "vocab_size": BERT_SPEC.vocab_size,
        "hidden_size": BERT_SPEC.hidden_size,
        "num_hidden_layers": BERT_SPEC.num_layers,
        "num_attention_heads": BERT_SPEC.num_heads,
        "intermediate_size": BERT_SPEC.intermediate_size,
        "max_position_embeddings": BERT_SPEC.max_position_embeddings,
        "type_vocab_size": BERT_SPEC.type_vocab_size,
        "layer_norm_eps": BERT_SPEC.layer_norm_eps,
        "id2label": {"0": "LABEL_0"}, "pad_token_id": 0,
    }
    (tmp_path / "config.json").write_text(json.dumps(cfgd))
    # save under HF names so load_dir's mapping is exercised
    from transformers import BertConfig, BertForSequenceClassification

    hf = BertForSequenceClassification(BertConfig(
        vocab_size=BERT_SPEC.vocab_size, hidden_size=BERT_SPEC.hidden_size,
        num_hidden_layers=BERT_SPEC.num_layers,
        num_attention_heads=BERT_SPEC.num_heads,
        intermediate_size=BERT_SPEC.intermediate_size,
        max_position_embeddings=BERT_SPEC.max_position_embeddings,
        type_vocab_size=BERT_SPEC.type_vocab_size, num_labels=1,
        pad_token_id=0)).eval().float()
    save_file({k: v.contiguous() for k, v in hf.state_dict().items()
               if "position_ids" not in k},
              str(tmp_path / "model.safetensors"))
    m.load_hf_state_dict(hf.state_dict())

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-rerank", "--source", "local",
        "--model-ref", str(tmp_path), "--port", str(port), "--device", "cpu",
    ])
    try:
        t0 = time.time()
        while time.time() - t0 < 90:
            if proc.poll() is not None:
                raise AssertionError(f"server exited {proc.returncode}")
            try:
                r = httpx.get(f"http://127.0.0.1:{port}/health", timeout=2)
                if r.status_code == 200:
                    assert r.json()["mode"] == "reranker"
                    break
            except httpx.HTTPError:
                pass
            time.sleep(0.5)
        else:
            raise AssertionError("encoder server never became healthy")

        docs = ["deep learning on MI355X", "cooking pasta", "gpu kernels"]
        r = httpx.post(f"http://127.0.0.1:{port}/v1/rerank", json={
            "model": "tiny-rerank", "query": "gpu compute", "documents": docs,
            "top_n": 2}, timeout=60)
        assert r.status_code == 200, r.text
        res = r.json()["results"]
        assert len(res) == 2
        assert res[0]["relevance_score"] >= res[1]["relevance_score"]
        assert all(0.0 < x["relevance_score"] < 1.0 for x in res)

        # wire scores == in-process runner scores (same tokenizer+weights)
        from gpustack_amd.worker.engine_server import load_tokenizer

        tok = load_tokenizer(str(tmp_path), BERT_SPEC.vocab_size)

        def enc(t):
            ids = tok.encode(t)
            return list(ids.ids if hasattr(ids, "ids") else ids) or [0]

        runner = CrossEncoderRunner(m, cls_id=101, sep_id=102)
        want = runner.score(enc("gpu compute"), [enc(d) for d in docs])
        got = {x["index"]: x["relevance_score"]
               for x in httpx.post(
                   f"http://127.0.0.1:{port}/v1/rerank",
                   json={"model": "tiny-rerank", "query": "gpu compute",
                         "documents": docs}, timeout=60).json()["results"]}
        for i, w in enumerate(want):
            assert abs(got[i] - 1.0 / (1.0 + math.exp(-w))) < 1e-5

        r = httpx.post(f"http://127.0.0.1:{port}/v1/score", json={
            "model": "tiny-rerank", "text_1": "gpu compute",
            "text_2": docs[:2]}, timeout=60)
        assert len(r.json()["data"]) == 2
        r = httpx.get(f"http://127.0.0.1:{port}/metrics", timeout=10)
        assert "gpustack_rerank_pairs_total" in r.text
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_model_registry_categories():
    from gpustack_amd.utils.model_registry import (
        categories_for_architecture, categories_for_model,
    )

    assert categories_for_architecture(
        "XLMRobertaForSequenceClassification") == ["reranker"]
    assert categories_for_architecture("BertModel") == ["embedding"]
    assert categories_for_architecture("LlamaForCausalLM") == ["llm"]
    assert categories_for_architecture(
        "DeepseekV3ForCausalLM") == ["llm", "moe"]
    assert categories_for_architecture(
        "WhisperForConditionalGeneration") == ["speech_to_text"]
    assert categories_for_model("preset", "deepseek-v3") == ["llm", "moe"]
    assert categories_for_model("hf", "org/unknown-remote") is None


def test_reranker_checkpoint_auto_categorized_and_sized(tmp_path):
    """A reranker dir: /v2/models create derives categories=[reranker];
    the scheduler sizing spec resolves (no 'cannot resolve model spec')."""
    import json as _json

    from gpustack_amd.scheduler.policies import model_spec_for

    (tmp_path / "config.json").write_text(_json.dumps({
        "architectures": ["XLMRobertaForSequenceClassification"],
        "vocab_size": 250002, "hidden_size": 1024,
        "num_hidden_layers": 24, "num_attention_heads": 16,
        "intermediate_size": 4096, "max_position_embeddings": 8194,
        "type_vocab_size": 1, "pad_token_id": 1,
    }))
    spec = model_spec_for({"source": "local", "model_ref": str(tmp_path)})
    assert spec is not None and spec.hidden_size == 1024
    assert 1 << 30 < spec.weight_bytes() < 4 << 30  # ~1.1 GB class model

    import tempfile

    from starlette.testclient import TestClient

    from gpustack_amd.config import Config
    from gpustack_amd.server.app import create_app

    cfg = Config(data_dir=tempfile.mkdtemp(), bootstrap_password="pw123")
    app = create_app(cfg, start_background=False)
    client = TestClient(app)
    r = client.post("/auth/login", json={"username": "admin",
                                         "password": "pw123"})
    client.headers["Authorization"] = f"Bearer {r.json()['token']}"
    r = client.post("/v2/models", json={
        "name": "bge-reranker", "source": "local",
        "model_ref": str(tmp_path)})
    assert r.status_code == 201, r.text
    assert r.json()["categories"] == ["reranker"]
