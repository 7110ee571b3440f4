"""Weights-path deploys: model = local HF checkpoint directory
(config.json + *.safetensors + tokenizer.json). The reference's "image"
analog (SURVEY.md §2.1 docker-client row): deploy validates the path,
attach loads weights and the REAL tokenizer (BPE via `tokenizers`).
"""

import json
import os

import pytest
import torch

from agentainer_amd.config import load_config
from agentainer_amd.engine.llm import LLMEngine
from agentainer_amd.engine.tokenizer import HFTokenizer
from agentainer_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM, config_from_hf
from agentainer_amd.service import Runtime
from agentainer_amd.store import Store


def _write_checkpoint(d, train_tokenizer=True):
    """Export tiny-llama-shaped random weights as an HF checkpoint."""
    from safetensors.torch import save_file

    os.makedirs(d, exist_ok=True)
    # train a real byte-level BPE tokenizer on a tiny corpus (offline)
    vocab = 512
    if train_tokenizer:
        from tokenizers import Tokenizer, models, pre_tokenizers, trainers, decoders
        tok = Tokenizer(models.BPE(unk_token=None))
        tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
        tok.decoder = decoders.ByteLevel()
        trainer = trainers.BpeTrainer(
            vocab_size=vocab, special_tokens=["<eos>"],
            initial_alphabet=pre_tokenizers.ByteLevel.alphabet())
        corpus = ["the agent replies tersely", "[system] you are helpful",
                  "[user] hello there [assistant] hi"] * 20
        tok.train_from_iterator(corpus, trainer)
        tok.save(os.path.join(d, "tokenizer.json"))
        vocab = tok.get_vocab_size()
    cfg = {
        "model_type": "llama", "architectures": ["LlamaForCausalLM"],
        "vocab_size": max(vocab, 512), "hidden_size": 512,
        "num_hidden_layers": 2, "num_attention_heads": 4,
        "num_key_value_heads": 1, "head_dim": 128,
        "intermediate_size": 1024, "rope_theta": 500000.0,
        "max_position_embeddings": 4096, "rms_norm_eps": 1e-5,
        "tie_word_embeddings": True, "eos_token_id": 0,
    }
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    torch.manual_seed(7)
    lc = config_from_hf(d)
    m = LlamaForCausalLM(lc, device="cpu", seed=7)
    sd = {"model.embed_tokens.weight": m.embed.data.clone(),
          "model.norm.weight": m.final_ln.data.clone()}
    for i, layer in enumerate(m.layers):
        pfx = f"model.layers.{i}."
        qkv = layer.attn.qkv_proj.data
        q_sz, kv_sz = lc.q_size, lc.kv_size
        sd[pfx + "self_attn.q_proj.weight"] = qkv[:q_sz].clone()
        sd[pfx + "self_attn.k_proj.weight"] = qkv[q_sz:q_sz + kv_sz].clone()
        sd[pfx + "self_attn.v_proj.weight"] = qkv[q_sz + kv_sz:].clone()
        sd[pfx + "self_attn.o_proj.weight"] = layer.attn.o_proj.data.clone()
        gu = layer.mlp.gate_up.data
        sd[pfx + "mlp.gate_proj.weight"] = gu[:lc.intermediate_size].clone()
        sd[pfx + "mlp.up_proj.weight"] = gu[lc.intermediate_size:].clone()
        sd[pfx + "mlp.down_proj.weight"] = layer.mlp.down.data.clone()
        sd[pfx + "input_layernorm.weight"] = layer.input_ln.data.clone()
        sd[pfx + "post_attention_layernorm.weight"] = layer.post_ln.data.clone()
    save_file(sd, os.path.join(d, "model.safetensors"))
    return lc


def test_config_from_hf_mapping(tmp_path):
    d = str(tmp_path / "ckpt")
    lc = _write_checkpoint(d, train_tokenizer=False)
    assert lc.n_layers == 2 and lc.n_heads == 4 and lc.n_kv_heads == 1
    assert lc.head_dim == 128 and lc.tie_embeddings is True
    with pytest.raises(ValueError):
        cfgp = os.path.join(d, "config.json")
        c = json.load(open(cfgp))
        c["model_type"] = "gpt_bigcode"
        c["architectures"] = ["GPTBigCodeForCausalLM"]
        json.dump(c, open(cfgp, "w"))
        config_from_hf(d)


def test_deploy_by_weights_path(tmp_path):
    d = str(tmp_path / "ckpt")
    _write_checkpoint(d)
    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / "root")
    cfg.data["store"]["path"] = root
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    s = Store(root + "/state", sync="interval")
    rt = Runtime(cfg, engine=LLMEngine(s, cfg, device="cpu", state_root=root),
                 store=s, state_root=root)
    try:
        a = rt.agents.deploy(name="ckpt-agent", model=d,
                             sampling={"max_tokens": 6})
        rt.agents.start(a.id)
        inst = rt.engine._instances[d]
        assert isinstance(inst.tokenizer, HFTokenizer)
        assert inst.tokenizer.eos_id == 0  # from config.json
        # real tokenizer round trip
        ids = inst.tokenizer.encode("the agent replies")
        assert ids and inst.tokenizer.decode(ids) == "the agent replies"
        st, p = rt.agent_request(a.id, "POST", "/chat",
                                 body={"message": "hello there"})
        assert st == 200
        assert 0 < p["tokens"] <= 6
        # streamed deltas concatenate to the blocking-style response
        events = list(rt.engine.chat_stream(a.id, "hello there"))
        assert events[-1]["done"] is True
        text = "".join(e["text"] for e in events[:-1])
        assert text == events[-1]["response"]
    finally:
        rt.shutdown()


def test_deploy_unknown_model_rejected(tmp_path):
    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / "root")
    cfg.data["store"]["path"] = root
    s = Store(root + "/state", sync="interval")
    rt = Runtime(cfg, engine=LLMEngine(s, cfg, device="cpu", state_root=root),
                 store=s, state_root=root)
    try:
        from agentainer_amd.engine.base import ModelNotFound
        with pytest.raises(ModelNotFound):
            rt.agents.deploy(name="bad", model="no-such-model-xyz")
    finally:
        rt.shutdown()


def test_deploy_qwen2_checkpoint(tmp_path):
    """Qwen2-family weights-path deploy: qkv biases loaded and applied."""
    from safetensors.torch import save_file

    d = str(tmp_path / "qwen")
    os.makedirs(d, exist_ok=True)
    cfg = {
        "model_type": "qwen2", "architectures": ["Qwen2ForCausalLM"],
        "vocab_size": 512, "hidden_size": 512, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 1, "head_dim": 128,
        "intermediate_size": 1024, "rope_theta": 1e6,
        "max_position_embeddings": 4096, "rms_norm_eps": 1e-6,
        "tie_word_embeddings": True, "eos_token_id": 0,
    }
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    lc = config_from_hf(d)
    assert lc.qkv_bias is True and lc.rope_theta == 1e6
    torch.manual_seed(9)
    m = LlamaForCausalLM(lc, device="cpu", seed=9)
    sd = {"model.embed_tokens.weight": m.embed.data.clone(),
          "model.norm.weight": m.final_ln.data.clone()}
    q_sz, kv_sz = lc.q_size, lc.kv_size
    for i, layer in enumerate(m.layers):
        pfx = f"model.layers.{i}."
        qkv = layer.attn.qkv_proj.data
        sd[pfx + "self_attn.q_proj.weight"] = qkv[:q_sz].clone()
        sd[pfx + "self_attn.k_proj.weight"] = qkv[q_sz:q_sz + kv_sz].clone()
        sd[pfx + "self_attn.v_proj.weight"] = qkv[q_sz + kv_sz:].clone()
        bias = torch.randn(q_sz + 2 * kv_sz, dtype=torch.bfloat16) * 0.02
        sd[pfx + "self_attn.q_proj.bias"] = bias[:q_sz].clone()
        sd[pfx + "self_attn.k_proj.bias"] = bias[q_sz:q_sz + kv_sz].clone()
        sd[pfx + "self_attn.v_proj.bias"] = bias[q_sz + kv_sz:].clone()
        sd[pfx + "self_attn.o_proj.weight"] = layer.attn.o_proj.data.clone()
        gu = layer.mlp.gate_up.data
        sd[pfx + "mlp.gate_proj.weight"] = gu[:lc.intermediate_size].clone()
        sd[pfx + "mlp.up_proj.weight"] = gu[lc.intermediate_size:].clone()
        sd[pfx + "mlp.down_proj.weight"] = layer.mlp.down.data.clone()
        sd[pfx + "input_layernorm.weight"] = layer.input_ln.data.clone()
        sd[pfx + "post_attention_layernorm.weight"] = layer.post_ln.data.clone()
    save_file(sd, os.path.join(d, "model.safetensors"))

    cfg2 = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / "root")
    cfg2.data["store"]["path"] = root
    cfg2.data["engine"]["kv_pool_gb"] = 0.01
    s = Store(root + "/state", sync="interval")
    rt = Runtime(cfg2, engine=LLMEngine(s, cfg2, device="cpu", state_root=root),
                 store=s, state_root=root)
    try:
        a = rt.agents.deploy(name="qwen-agent", model=d,
                             sampling={"max_tokens": 5})
        rt.agents.start(a.id)
        inst = rt.engine._instances[d]
        # biases survived the load (nonzero)
        b0 = inst.model.layers[0].attn.qkv_bias
        assert b0 is not None and b0.abs().sum() > 0
        st, p = rt.agent_request(a.id, "POST", "/chat",
                                 body={"message": "ni hao"})
        assert st == 200 and 0 < p["tokens"] <= 5
    finally:
        rt.shutdown()
