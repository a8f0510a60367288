"""Graph-captured GPT-2 decode for RLHF rollouts on MI355X.

The HF single-token decode path can't be hipGraph-captured on this
stack (``StaticCache`` + ROCm 7.2 segfaults, and ``generate`` burns
milliseconds of host logic per token), so :class:`GraphedGPT2Decoder`
reimplements the GPT-2 decode step directly on the model's own
parameters with manually-managed static KV buffers: one prefill
(ordinary HF forward) and then ONE captured graph replay per generated
token — embedding lookup, 12 blocks of LN/attention-over-static-KV/MLP,
final LN, tied-lm-head logits, multinomial sampling and all buffer
bookkeeping.

The captured kernels read the policy's bf16 weight CACHES (refreshed
once per training iteration via :meth:`refresh_weights`), so the same
graph keeps decoding as training updates the fp32 masters.

Reference analog: torchrl's vLLM-backed generation
(torchrl/modules/llm/backends/vllm*); here the engine is a hand-rolled
static-KV loop because the image has no vLLM and HF's static path dies.
Numerics validated against the eager HF forward in
tests/test_llm_bench.py (same sampling distribution as
``generate(do_sample=True, top_k=0)``).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

__all__ = ["GraphedGPT2Decoder"]


class GraphedGPT2Decoder:
    def __init__(self, model, batch: int, max_len: int, device,
                 dtype=torch.bfloat16):
        self.model = model
        self.cfg = model.config
        self.B = batch
        self.T = max_len
        self.device = device
        self.dtype = dtype
        c = self.cfg
        self.L, self.nh, self.E = c.n_layer, c.n_head, c.n_embd
        self.dh = self.E // self.nh
        self.V = c.vocab_size
        tr = model.transformer
        self._blocks = list(tr.h)
        self._tr = tr
        # bf16 weight caches (refreshed per training iteration)
        self._w = {}
        self.refresh_weights()
        # static KV + bookkeeping buffers
        kv = (self.L, self.B, self.nh, self.T, self.dh)
        self.K = torch.zeros(kv, device=device, dtype=dtype)
        self.Vv = torch.zeros(kv, device=device, dtype=dtype)
        self.mask = torch.full((self.T,), float("-inf"), device=device)
        self.cur = torch.zeros(self.B, 1, dtype=torch.long, device=device)
        self.pos = torch.zeros(1, dtype=torch.long, device=device)
        self.pos_out = torch.zeros(1, dtype=torch.long, device=device)
        self.gen_buf = torch.zeros(self.B, self.T, dtype=torch.long,
                                   device=device)
        self._graph = None

    # ------------------------------------------------------------------ #
    def refresh_weights(self) -> None:
        """Re-cast the decode weight caches from the live parameters
        (call once per training iteration, after the optimizer step)."""
        bf = self.dtype
        w = self._w
        tr = self._tr
        first = not w
        def put(name, t, dtype):
            t = t.detach().to(dtype)
            if first:
                w[name] = t.clone()
            else:
                w[name].copy_(t)
        put("wte", tr.wte.weight, bf)
        put("wpe", tr.wpe.weight, bf)
        put("lnf_w", tr.ln_f.weight, torch.float32)
        put("lnf_b", tr.ln_f.bias, torch.float32)
        for i, blk in enumerate(self._blocks):
            put(f"ln1_w{i}", blk.ln_1.weight, torch.float32)
            put(f"ln1_b{i}", blk.ln_1.bias, torch.float32)
            put(f"attn_w{i}", blk.attn.c_attn.weight, bf)   # [E, 3E]
            put(f"attn_b{i}", blk.attn.c_attn.bias, bf)
            put(f"proj_w{i}", blk.attn.c_proj.weight, bf)   # [E, E]
            put(f"proj_b{i}", blk.attn.c_proj.bias, bf)
            put(f"ln2_w{i}", blk.ln_2.weight, torch.float32)
            put(f"ln2_b{i}", blk.ln_2.bias, torch.float32)
            put(f"fc_w{i}", blk.mlp.c_fc.weight, bf)        # [E, 4E]
            put(f"fc_b{i}", blk.mlp.c_fc.bias, bf)
            put(f"fc2_w{i}", blk.mlp.c_proj.weight, bf)     # [4E, E]
            put(f"fc2_b{i}", blk.mlp.c_proj.bias, bf)

    # ------------------------------------------------------------------ #
    def _step_logits(self) -> torch.Tensor:
        """One decode forward over the static KV: [B, V] fp32 logits
        for the token ids in ``self.cur`` at position ``self.pos``."""
        w = self._w
        B, nh, dh, E = self.B, self.nh, self.dh, self.E
        eps = self.cfg.layer_norm_epsilon
        # the new token becomes attendable
        self.mask.index_fill_(0, self.pos, 0.0)
        x = (w["wte"].index_select(0, self.cur.reshape(-1))
             + w["wpe"].index_select(0, self.pos)).reshape(B, 1, E)
        scale = 1.0 / float(dh) ** 0.5
        for i in range(self.L):
            xn = F.layer_norm(x.float(), (E,), w[f"ln1_w{i}"],
                              w[f"ln1_b{i}"], eps).to(self.dtype)
            qkv = torch.addmm(w[f"attn_b{i}"], xn.reshape(B, E),
                              w[f"attn_w{i}"])
            q, k, v = qkv.split(E, dim=-1)
            q = q.reshape(B, nh, 1, dh)
            k = k.reshape(B, 1, nh, dh).transpose(1, 2)
            v = v.reshape(B, 1, nh, dh).transpose(1, 2)
            self.K[i].index_copy_(2, self.pos, k)
            self.Vv[i].index_copy_(2, self.pos, v)
            att = torch.matmul(q, self.K[i].transpose(-1, -2)).float()
            att = att * scale + self.mask
            probs = att.softmax(-1).to(self.dtype)
            o = torch.matmul(probs, self.Vv[i])      # [B, nh, 1, dh]
            o = o.transpose(1, 2).reshape(B, E)
            x = x + torch.addmm(w[f"proj_b{i}"], o,
                                w[f"proj_w{i}"]).reshape(B, 1, E)
            xn = F.layer_norm(x.float(), (E,), w[f"ln2_w{i}"],
                              w[f"ln2_b{i}"], eps).to(self.dtype)
            h = torch.addmm(w[f"fc_b{i}"], xn.reshape(B, E), w[f"fc_w{i}"])
            h = F.gelu(h, approximate="tanh")
            x = x + torch.addmm(w[f"fc2_b{i}"], h,
                                w[f"fc2_w{i}"]).reshape(B, 1, E)
        xn = F.layer_norm(x.float(), (E,), w["lnf_w"], w["lnf_b"], eps)
        logits = torch.matmul(xn.to(self.dtype).reshape(B, E),
                              w["wte"].t())
        return logits.float()

    def _decode_body(self) -> None:
        logits = self._step_logits()
        nxt = torch.multinomial(torch.softmax(logits, -1), 1)
        self.cur.copy_(nxt)
        self.gen_buf.index_copy_(1, self.pos_out, nxt)
        self.pos.add_(1)
        self.pos_out.add_(1)

    # ------------------------------------------------------------------ #
    def prefill(self, prompts: torch.Tensor) -> None:
        """HF forward over the prompts; copies its KV into the static
        buffers and samples the first generated token."""
        P = prompts.shape[1]
        with torch.no_grad(), torch.autocast(
            "cuda", dtype=self.dtype, cache_enabled=False,
            enabled=prompts.is_cuda,
        ):
            out = self.model(input_ids=prompts, use_cache=True)
        past = out.past_key_values
        for i in range(self.L):
            if hasattr(past, "layers"):  # transformers >= 4.54 Cache
                k, v = past.layers[i].keys, past.layers[i].values
            elif hasattr(past, "key_cache"):
                k, v = past.key_cache[i], past.value_cache[i]
            else:
                k, v = past[i]
            self.K[i, :, :, :P] = k.to(self.dtype)
            self.Vv[i, :, :, :P] = v.to(self.dtype)
        self.mask.fill_(float("-inf"))
        self.mask[:P] = 0.0
        logits = out.logits[:, -1].float()
        first = torch.multinomial(torch.softmax(logits, -1), 1)
        self.cur.copy_(first)
        self.gen_buf[:, 0:1] = first
        self.pos.fill_(P)
        self.pos_out.fill_(1)

    def capture(self) -> bool:
        """hipGraph-capture the decode step (after at least one
        prefill).  Returns False (eager fallback stays) on failure."""
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side), torch.no_grad():
                for _ in range(3):
                    self._decode_body()
            torch.cuda.current_stream().wait_stream(side)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g), torch.no_grad():
                self._decode_body()
            self._graph = g
            return True
        except Exception:
            self._graph = None
            return False

    def rollout(self, prompts: torch.Tensor, gen_tokens: int) -> torch.Tensor:
        """Generate ``gen_tokens`` tokens; returns [B, P + G] ids."""
        self.prefill(prompts)
        with torch.no_grad():
            for _ in range(gen_tokens - 1):
                if self._graph is not None:
                    self._graph.replay()
                else:
                    self._decode_body()
        return torch.cat(
            [prompts, self.gen_buf[:, :gen_tokens]], dim=1
        )
