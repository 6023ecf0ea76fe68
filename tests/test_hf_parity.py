"""HF-parity golden tests: map transformers Qwen3/Qwen3MoE weights through the
state mappers and compare logits (reference: test/d9d_test/.../test_hf.py)."""

import pytest
import torch
import torch.nn.functional as F


def _kl(p_logits, q_logits):
    p = F.log_softmax(p_logits.float(), dim=-1)
    q = F.log_softmax(q_logits.float(), dim=-1)
    return F.kl_div(q, p, log_target=True, reduction="batchmean").item()


@pytest.mark.filterwarnings("ignore")
def test_hf_parity_qwen3_dense():
    from transformers import Qwen3Config, Qwen3ForCausalLM

    from d9d_amd.model_state.io import _StreamingApplier  # noqa: F401
    from d9d_amd.module.model.qwen3_dense import (
        Qwen3DenseForCausalLM,
        Qwen3DenseModelParameters,
    )
    from d9d_amd.module.model.qwen3_dense.huggingface import hf_to_d9d_mapper

    p = Qwen3DenseModelParameters(
        hidden_size=64,
        intermediate_size=128,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=16,
        num_hidden_layers=2,
        rope_base=10000.0,
        split_vocab_size={"regular": 480, "special": 32},
    )
    hf_cfg = Qwen3Config(
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        rms_norm_eps=1e-6,
        attention_bias=False,
        tie_word_embeddings=False,
    )
    torch.manual_seed(0)
    hf = Qwen3ForCausalLM(hf_cfg).eval()

    ours = Qwen3DenseForCausalLM(p).eval()
    mapped = hf_to_d9d_mapper(p).apply(dict(hf.state_dict()))
    missing = set(ours.state_dict()) - set(mapped)
    assert not missing, f"unmapped keys: {sorted(missing)[:8]}"
    ours.load_state_dict(mapped)

    ids = torch.randint(0, 512, (2, 24))
    with torch.no_grad():
        hf_logits = hf(ids).logits
        h = ours.model(input_ids=ids)["hidden_states"]
        our_logits = ours.lm_head.logits(h)
    assert _kl(hf_logits, our_logits) < 1e-4
    # per-token logps path agrees with explicit logits
    with torch.no_grad():
        out = ours(input_ids=ids, labels=ids)
    ref_logp = torch.log_softmax(our_logits.float(), -1).gather(
        -1, ids.unsqueeze(-1)
    ).squeeze(-1)
    torch.testing.assert_close(out["logps"], ref_logp, rtol=1e-3, atol=1e-4)


@pytest.mark.filterwarnings("ignore")
def test_hf_parity_qwen3_moe():
    from transformers import Qwen3MoeConfig, Qwen3MoeForCausalLM

    from d9d_amd.module.model.qwen3_moe import (
        Qwen3MoEForCausalLM,
        Qwen3MoEModelParameters,
    )
    from d9d_amd.module.model.qwen3_moe.huggingface import (
        d9d_to_hf_mapper,
        hf_to_d9d_mapper,
    )

    p = Qwen3MoEModelParameters(
        hidden_size=64,
        intermediate_size=48,
        num_experts=8,
        experts_top_k=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=16,
        num_hidden_layers=2,
        rope_base=10000.0,
        split_vocab_size={"regular": 480, "special": 32},
    )
    hf_cfg = Qwen3MoeConfig(
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        moe_intermediate_size=48,
        num_experts=8,
        num_experts_per_tok=2,
        norm_topk_prob=True,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        decoder_sparse_step=1,
        mlp_only_layers=[],
        shared_expert_intermediate_size=0,
        tie_word_embeddings=False,
    )
    torch.manual_seed(0)
    hf = Qwen3MoeForCausalLM(hf_cfg).eval()

    ours = Qwen3MoEForCausalLM(p).eval()
    mapped = hf_to_d9d_mapper(p).apply(dict(hf.state_dict()))
    state_keys = {
        k for k in ours.state_dict() if "tokens_per_expert" not in k
    }
    missing = state_keys - set(mapped)
    assert not missing, f"unmapped keys: {sorted(missing)[:8]}"
    ours.load_state_dict(mapped, strict=False)

    ids = torch.randint(0, 512, (2, 16))
    with torch.no_grad():
        hf_logits = hf(ids).logits
        h = ours.model(input_ids=ids)["hidden_states"]
        our_logits = ours.lm_head.logits(h)
    assert _kl(hf_logits, our_logits) < 1e-4

    # round-trip back to HF format
    back = d9d_to_hf_mapper(p).apply(mapped)
    for k, v in hf.state_dict().items():
        torch.testing.assert_close(back[k], v, rtol=1e-6, atol=1e-7)


@pytest.mark.filterwarnings("ignore")
def test_hf_parity_llama3():
    from transformers import LlamaConfig, LlamaForCausalLM

    from d9d_amd.module.model.llama3 import (
        Llama3ForCausalLM,
        Llama3ModelParameters,
        hf_to_d9d_mapper,
    )

    p = Llama3ModelParameters(
        hidden_size=64,
        intermediate_size=128,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=16,
        num_hidden_layers=2,
        rope_base=10000.0,
        split_vocab_size={"regular": 480, "special": 32},
    )
    hf_cfg = LlamaConfig(
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        rms_norm_eps=1e-6,
        attention_bias=False,
        mlp_bias=False,
        tie_word_embeddings=False,
    )
    torch.manual_seed(0)
    hf = LlamaForCausalLM(hf_cfg).eval()

    ours = Llama3ForCausalLM(p).eval()
    mapped = hf_to_d9d_mapper(p).apply(dict(hf.state_dict()))
    missing = set(ours.state_dict()) - set(mapped)
    assert not missing, f"unmapped keys: {sorted(missing)[:8]}"
    ours.load_state_dict(mapped)

    ids = torch.randint(0, 512, (2, 24))
    with torch.no_grad():
        hf_logits = hf(ids).logits
        h = ours.model(input_ids=ids)["hidden_states"]
        our_logits = ours.lm_head.logits(h)
    assert _kl(hf_logits, our_logits) < 1e-4


@pytest.mark.filterwarnings("ignore")
def test_llama3_export_roundtrip():
    """d9d -> HF -> d9d mapping is the identity on Llama-3 state."""
    from d9d_amd.module.model.llama3 import (
        Llama3ForCausalLM,
        Llama3ModelParameters,
        d9d_to_hf_mapper,
        hf_to_d9d_mapper,
    )

    p = Llama3ModelParameters(
        hidden_size=32,
        intermediate_size=64,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=16,
        num_hidden_layers=2,
        split_vocab_size={"regular": 96, "special": 8},
    )
    torch.manual_seed(1)
    m = Llama3ForCausalLM(p)
    m.init_weights()
    sd = {k: v.clone() for k, v in m.state_dict().items()}
    hf_sd = d9d_to_hf_mapper(p).apply(dict(sd))
    assert "model.layers.0.self_attn.q_proj.weight" in hf_sd
    assert not any("q_norm" in k for k in hf_sd)
    back = hf_to_d9d_mapper(p).apply(hf_sd)
    assert set(back) == set(sd)
    for k in sd:
        torch.testing.assert_close(back[k], sd[k], msg=lambda m_: f"{k}: {m_}")
