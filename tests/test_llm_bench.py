"""GPU tests for the RLHF decode utilities."""
import pytest
import torch  # noqa: F401

pytest.importorskip("transformers")

@pytest.mark.gpu
def test_graphed_gpt2_decoder_matches_hf():
    """GraphedGPT2Decoder's manual decode step must reproduce the HF
    forward's next-token logits over the same KV state."""
    import torch
    from transformers import GPT2Config, GPT2LMHeadModel

    from rl_amd.modules.llm.decode import GraphedGPT2Decoder

    torch.manual_seed(0)
    cfg = GPT2Config(n_layer=4, n_head=4, n_embd=128, vocab_size=503)
    model = GPT2LMHeadModel(cfg).cuda().eval()
    B, P = 3, 9
    dec = GraphedGPT2Decoder(model, B, P + 8, "cuda")
    prompts = torch.randint(0, 503, (B, P), device="cuda")
    dec.prefill(prompts)
    tok = dec.cur.clone()
    logits = dec._step_logits()
    # HF reference: full forward over [prompts, tok]
    full = torch.cat([prompts, tok], 1)
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16,
                                         cache_enabled=False):
        ref = model(full).logits[:, -1].float()
    assert torch.allclose(logits, ref, atol=0.5, rtol=0.05), (
        (logits - ref).abs().max()
    )
    # top-1 agreement is what sampling cares about
    assert (logits.argmax(-1) == ref.argmax(-1)).float().mean() > 0.9
