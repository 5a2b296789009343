"""On-device sampling.

Analog of the reference's `modules/generation/sampling.py` (Sampler at :241,
greedy :372-388, multinomial :285-370, per-request params :97-198): sampling
runs inside the captured graph on vocab-sharded logits — greedy via
distributed argmax, multinomial via distributed top-`global_topk` then the
fused HIP top-k/top-p/temperature/inverse-CDF kernel.

Per-request `sampling_params` is a (B, 3) tensor [top_k, top_p, temperature]
exactly like the reference so serving stacks can drive it per sequence.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from neuronx_distributed_inference_amd import ops
from neuronx_distributed_inference_amd.config import OnDeviceSamplingConfig
from neuronx_distributed_inference_amd.parallel import ops as pops
from neuronx_distributed_inference_amd.parallel import state as ps


def prepare_sampling_params(
    batch_size: int,
    top_k=1,
    top_p: float = 1.0,
    temperature: float = 1.0,
    device=None,
) -> torch.Tensor:
    """Build the (B,3) sampling-params tensor (ref sampling.py:183-198)."""

    def _t(v, dtype):
        t = torch.as_tensor(v, dtype=dtype, device=device)
        if t.dim() == 0:
            t = t.expand(batch_size).clone()
        if t.numel() != batch_size:
            raise ValueError(f"sampling param wrong batch size: {t.numel()} != {batch_size}")
        return t.float()

    return torch.stack(
        [_t(top_k, torch.float32), _t(top_p, torch.float32), _t(temperature, torch.float32)],
        dim=1,
    )


def validate_sampling_params(params: torch.Tensor, config: OnDeviceSamplingConfig):
    if params.dim() != 2 or params.shape[1] != 3:
        raise ValueError(f"sampling_params must be (B,3), got {tuple(params.shape)}")
    if (params[:, 0] > config.global_topk).any():
        raise ValueError(f"top_k exceeds global_topk={config.global_topk}")
    if ((params[:, 1] <= 0) | (params[:, 1] > 1.0)).any():
        raise ValueError("top_p must be in (0, 1]")


class Sampler(nn.Module):
    """Samples token ids from vocab-sharded logits on device."""

    def __init__(self, config: OnDeviceSamplingConfig, vocab_size: int,
                 deterministic: Optional[bool] = None):
        super().__init__()
        self.config = config
        self.vocab_size = vocab_size
        self.deterministic = config.deterministic if deterministic is None else deterministic
        self._gen: Optional[torch.Generator] = None
        # The greedy/multinomial choice must be static (no host sync inside a
        # captured graph). The multinomial path handles per-row top_k=1
        # correctly (softmax over one candidate picks the argmax), so dynamic
        # per-request configs always take it.
        self.static_greedy = not (config.do_sample or config.dynamic)

    def _rand(self, batch: int, device) -> torch.Tensor:
        if self.deterministic:
            if self._gen is None or self._gen.device != torch.device(device):
                self._gen = torch.Generator(device=device)
                self._gen.manual_seed(0)
            return torch.rand(batch, device=device, generator=self._gen)
        return torch.rand(batch, device=device)

    def forward(
        self, logits: torch.Tensor, sampling_params: Optional[torch.Tensor] = None
    ) -> torch.Tensor:
        """logits: (B, S, V/rank) or (B, V/rank) — samples from the last
        position; returns (B,) token ids (global vocab indices)."""
        if logits.dim() == 3:
            logits = logits[:, -1, :]
        B = logits.shape[0]
        if self.static_greedy:
            # no sampling-params tensor needed (and building one would do a
            # host->device copy, which is illegal inside graph capture)
            return pops.distributed_argmax(logits, dim=-1)
        if sampling_params is None:
            sampling_params = prepare_sampling_params(
                B,
                self.config.top_k if self.config.do_sample else 1,
                self.config.top_p,
                self.config.temperature,
                device=logits.device,
            )
        top_k = sampling_params[:, 0]

        # multinomial: distributed top-global_topk then fused sampling
        k = min(self.config.global_topk, self.vocab_size)
        vals, idxs = pops.distributed_topk(logits, k, dim=-1)
        rand = self._rand(B, logits.device)
        # `vals` is sorted descending, so the fused kernel's internal top-k is
        # the identity and its output indexes the candidate list; map back to
        # global vocab ids through `idxs`.
        local = ops.topk_topp_sample(
            vals.float(), top_k, sampling_params[:, 1], sampling_params[:, 2],
            rand, global_topk=k,
        )
        return idxs.gather(1, local.view(-1, 1)).squeeze(1)


def create_sampler(config: OnDeviceSamplingConfig, vocab_size: int) -> Sampler:
    return Sampler(config, vocab_size)


class DataParallelSampler(nn.Module):
    """Batch-sharded sampling (ref DataParallelSampler, sampling.py:467-579):
    all-to-all reshards (full batch, vocab shard) -> (batch shard, wider
    vocab), local top-k, payload gather to full vocab candidates, final
    top-k + fused sample, then all-gather the batch back."""

    def __init__(self, config: OnDeviceSamplingConfig, vocab_size: int,
                 dp_degree: int):
        super().__init__()
        self.inner = Sampler(config, vocab_size)
        self.config = config
        self.dp = dp_degree
        self.vocab_size = vocab_size

    def forward(self, logits: torch.Tensor, sampling_params=None) -> torch.Tensor:
        from neuronx_distributed_inference_amd.parallel import mappings

        if logits.dim() == 3:
            logits = logits[:, -1, :]
        B = logits.shape[0]
        import torch.distributed as dist

        if self.dp <= 1 or not dist.is_initialized() or B % self.dp != 0:
            return self.inner(logits, sampling_params)
        # (B, V/tp) -> (B/dp, dp*V/tp): shard batch, widen local vocab.
        # After the reshard each rank holds the FULL vocab for its batch
        # shard, so sampling is purely local (no distributed argmax/topk).
        resharded = mappings.all_to_all(logits, scatter_dim=0, gather_dim=1)
        rank = dist.get_rank(group=mappings.ps.get_tensor_model_parallel_group())
        per = B // self.dp
        if self.inner.static_greedy:
            local = resharded.argmax(dim=-1)
        else:
            sp = sampling_params
            if sp is None:
                sp = prepare_sampling_params(
                    B, self.config.top_k if self.config.do_sample else 1,
                    self.config.top_p, self.config.temperature,
                    device=logits.device,
                )
            sp = sp[(rank % self.dp) * per:((rank % self.dp) + 1) * per]
            k = min(self.config.global_topk, resharded.shape[-1])
            vals, idxs = resharded.topk(k, dim=-1)
            rand = self.inner._rand(per, logits.device)
            pick = ops.topk_topp_sample(
                vals.float(), sp[:, 0], sp[:, 1], sp[:, 2], rand, global_topk=k
            )
            local = idxs.gather(1, pick.view(-1, 1)).squeeze(1)
        # all-gather the batch shards back (every rank needs the full batch)
        return mappings.gather_along_dim(local.view(-1, 1), dim=0).view(-1)


# ----------------------------------------------------------------------
# Rejection sampling for speculative decoding (ref model_base.py:1678-1759:
# `_adjust_target_probs`, `_speculative_mask`, `_speculative_token_selection`
# and the on-device rand accept draw at :1706). All tensor ops, graph-safe.


def filtered_probs(vals: torch.Tensor, sampling_params: torch.Tensor) -> torch.Tensor:
    """Probabilities after temperature/top-k/top-p over sorted-descending
    top-`Kg` logits `vals` (B, S, Kg). `sampling_params` (B,3) =
    [top_k, top_p, temperature]. Mirrors the fused sampling kernel's
    filtering: temperature -> top-k mask (vals sorted, so a position
    cutoff) -> softmax -> top-p nucleus (keep tokens whose PRECEDING
    cumulative mass < top_p; the first is always kept) -> renormalize."""
    B, S, Kg = vals.shape
    tk = sampling_params[:, 0].view(B, 1, 1)
    tp = sampling_params[:, 1].view(B, 1, 1)
    temp = sampling_params[:, 2].view(B, 1, 1).clamp(min=1e-6)
    x = vals.float() / temp
    pos = torch.arange(Kg, device=vals.device).view(1, 1, Kg)
    x = x.masked_fill(pos >= tk, float("-inf"))
    probs = torch.softmax(x, dim=-1)
    cum = probs.cumsum(-1)
    keep = (cum - probs) < tp
    probs = probs * keep
    return probs / probs.sum(-1, keepdim=True).clamp(min=1e-30)


def _select_value(from_idx: torch.Tensor, from_val: torch.Tensor,
                  ids: torch.Tensor) -> torch.Tensor:
    """value of token `ids` (B,k) under the (indices, values) distribution
    (B,k,Kg); 0 where absent (ref `_select_from`, model_base.py:1664)."""
    eq = from_idx == ids.unsqueeze(-1)
    return (from_val * eq).sum(-1)


def _map_probs(from_idx: torch.Tensor, from_val: torch.Tensor,
               to_idx: torch.Tensor) -> torch.Tensor:
    """re-express (from_idx, from_val) probs on the `to_idx` support:
    out[b,j,i] = from_val at token to_idx[b,j,i] (0 if absent)."""
    eq = to_idx.unsqueeze(-1) == from_idx.unsqueeze(-2)  # (B,k,Kt,Kf)
    return (from_val.unsqueeze(-2) * eq).sum(-1)


def sample_from_probs(probs: torch.Tensor, rand: torch.Tensor) -> torch.Tensor:
    """Inverse-CDF sample: probs (..., Kg) need not be normalized; rand
    (...) in [0,1). Returns LOCAL candidate indices (...)."""
    cum = probs.cumsum(-1)
    total = cum[..., -1:]
    return (cum < rand.unsqueeze(-1) * total).sum(-1).clamp(max=probs.shape[-1] - 1)


class RejectionSampler(nn.Module):
    """Speculative accept/resample with true rejection sampling so spec
    decode works under top-k/top-p sampling (ref model_base.py:1678-1759).

    Contract: draft token d_j was sampled from q_j; accept d_j iff
    rand < min(1, p_j(d_j)/q_j(d_j)); at the first rejection resample from
    normalize(clamp(p_j - q_j, 0)); if all k accepted, sample the bonus
    from p_k. Output distribution == target distribution regardless of
    the draft (the speculative-sampling theorem), which the CPU oracle
    test checks empirically."""

    def __init__(self, config: OnDeviceSamplingConfig, vocab_size: int,
                 deterministic: Optional[bool] = None):
        super().__init__()
        self.config = config
        self.k_g = min(config.global_topk, vocab_size)
        self.deterministic = (
            config.deterministic if deterministic is None else deterministic
        )
        self._gen: Optional[torch.Generator] = None

    def _rand(self, *shape, device) -> torch.Tensor:
        if self.deterministic:
            if self._gen is None or self._gen.device != torch.device(device):
                self._gen = torch.Generator(device=device)
                self._gen.manual_seed(0)
            return torch.rand(*shape, device=device, generator=self._gen)
        return torch.rand(*shape, device=device)

    def dist(self, logits_shard: torch.Tensor,
             sampling_params: torch.Tensor):
        """(B,S,V/tp) vocab-sharded logits -> filtered probs + GLOBAL
        vocab indices (B,S,Kg), sorted descending."""
        vals, idxs = pops.distributed_topk(logits_shard, self.k_g, dim=-1)
        return filtered_probs(vals, sampling_params), idxs

    def sample(self, probs: torch.Tensor, idxs: torch.Tensor,
               rand: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Sample token ids (B,S) from filtered probs/indices (B,S,Kg)."""
        if rand is None:
            rand = self._rand(*probs.shape[:-1], device=probs.device)
        local = sample_from_probs(probs, rand)
        return idxs.gather(-1, local.unsqueeze(-1)).squeeze(-1)

    def accept(self, draft_ids, q_probs, q_idx, p_probs, p_idx,
               pad_id: int = -1):
        """draft_ids (B,k); q_* (B,k,Kg) draft dists; p_* (B,k+1,Kg)
        target dists. Returns (tokens (B,k+1) `pad_id`-padded after the
        first rejection's replacement, n_accept (B,))."""
        B, k = draft_ids.shape
        device = draft_ids.device
        p_head, p_head_idx = p_probs[:, :k], p_idx[:, :k]
        p_d = _select_value(p_head_idx, p_head, draft_ids)
        q_d = _select_value(q_idx, q_probs, draft_ids)
        ratio = (p_d / q_d.clamp(min=1e-30)).clamp(max=1.0)
        rand_acc = self._rand(B, k, device=device)
        acc = (rand_acc < ratio).long()
        acc_prefix = acc.cumprod(dim=1)                      # contiguous prefix
        n_accept = acc_prefix.sum(dim=1)                     # (B,)

        # adjusted residual distribution on the target support
        q_on_p = _map_probs(q_idx, q_probs, p_head_idx)      # (B,k,Kg)
        adj = (p_head - q_on_p).clamp(min=0)
        s = adj.sum(-1, keepdim=True)
        is_zero = s < 1e-30
        # ref: where the residual vanishes (q covers p exactly) fall back
        # to a uniform over the support (ref sets all-ones)
        adj = torch.where(is_zero, torch.ones_like(adj), adj / s.clamp(min=1e-30))
        adj = torch.cat([adj, p_probs[:, k:]], dim=1)        # (B,k+1,Kg)
        rand_res = self._rand(B, k + 1, device=device)
        local = sample_from_probs(adj, rand_res)
        res_ids = p_idx.gather(-1, local.unsqueeze(-1)).squeeze(-1)  # (B,k+1)

        idxpos = torch.arange(k + 1, device=device).view(1, -1)
        acc_mask = torch.cat(
            [acc_prefix.bool(),
             torch.zeros(B, 1, dtype=torch.bool, device=device)], dim=1)
        draft_ext = torch.cat(
            [draft_ids, torch.zeros(B, 1, dtype=draft_ids.dtype, device=device)],
            dim=1)
        tokens = torch.where(acc_mask, draft_ext, res_ids)
        keep = idxpos <= n_accept.view(B, 1)
        tokens = torch.where(keep, tokens, torch.full_like(tokens, pad_id))
        return tokens, n_accept
