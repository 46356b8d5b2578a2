"""Diffusion serving: DDIM sampling over the native DiT, with
distrifusion-style patch parallelism
(reference: colossalai/inference/core/diffusion_engine.py +
inference/modeling/layers/distrifusion.py).

``DiffusionEngine.generate`` runs classifier-free-guided DDIM. Patch
parallelism splits the patch sequence across an SP group; every block's
attention needs the FULL K/V, so each rank all-gathers them. After
``warmup_steps`` denoising steps the gather goes ASYNC: the attention
consumes the PREVIOUS step's gathered K/V for the remote shards (the
displaced-patch trick — adjacent denoising steps are nearly identical,
so stale remote context costs little quality) while this step's gather
rides under compute on the xGMI links.
"""

from typing import List, Optional

import torch
import torch.distributed as dist

from ..models.dit import DiT

__all__ = ["DiffusionEngine", "PatchParallelDiT", "ddim_sample"]


@torch.no_grad()
def ddim_sample(model, shape, steps: int = 50, guidance_scale: float = 4.0,
                labels: Optional[torch.Tensor] = None, eta: float = 0.0,
                generator: Optional[torch.Generator] = None, device=None):
    """Plain DDIM over a linear alpha-bar schedule; model predicts eps."""
    device = device or next(model.parameters()).device
    B = shape[0]
    x = torch.randn(*shape, generator=generator, device=device,
                    dtype=next(model.parameters()).dtype)
    T = 1000
    betas = torch.linspace(1e-4, 0.02, T, device=device)
    abar = torch.cumprod(1 - betas, dim=0)
    ts = torch.linspace(T - 1, 0, steps, device=device).long()
    cfg = model.model.config if hasattr(model, "model") else model.config
    use_cfg = guidance_scale != 1.0 and labels is not None
    if use_cfg:
        # batch cond + uncond in ONE forward (keeps the distrifusion
        # per-forward K/V caches consistent between the two branches)
        null = torch.full_like(labels, cfg.num_classes)
        labels_full = torch.cat([labels, null])
    for i, t in enumerate(ts):
        tb = t.expand(B)
        if use_cfg:
            eps2 = model(torch.cat([x, x]), torch.cat([tb, tb]), labels_full)["sample"]
            eps_c, eps_u = eps2.chunk(2)
            eps = eps_u + guidance_scale * (eps_c - eps_u)
        else:
            eps = model(x, tb, labels)["sample"]
        a_t = abar[t]
        a_prev = abar[ts[i + 1]] if i + 1 < steps else torch.ones((), device=device)
        x0 = (x - (1 - a_t).sqrt() * eps) / a_t.sqrt()
        x = a_prev.sqrt() * x0 + (1 - a_prev).sqrt() * eps
    return x


class PatchParallelDiT(torch.nn.Module):
    """Distrifusion wrapper: rank r owns patch shard r; attention K/V are
    all-gathered (sync during warmup, then async-displaced)."""

    def __init__(self, model: DiT, group, warmup_steps: int = 2):
        super().__init__()
        self.model = model
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        self.warmup_steps = warmup_steps
        self._step = 0
        self._kv_cache = [None] * len(model.blocks)   # last gathered K/V per block
        self._pending = [None] * len(model.blocks)    # (work, buffers) in flight

    def reset(self):
        self._step = 0
        self._kv_cache = [None] * len(self.model.blocks)
        self._pending = [None] * len(self.model.blocks)

    def _gather_kv(self, i, k_loc, v_loc, asynchronous):
        kv = torch.cat([k_loc, v_loc], dim=0).contiguous()
        bufs = [torch.empty_like(kv) for _ in range(self.world)]
        work = dist.all_gather(bufs, kv, group=self.group, async_op=asynchronous)
        return work, bufs

    @staticmethod
    def _assemble(bufs):
        ks, vs = [], []
        for b in bufs:
            kb, vb = b.chunk(2, dim=0)
            ks.append(kb)
            vs.append(vb)
        return torch.cat(ks, dim=1), torch.cat(vs, dim=1)

    def forward(self, latents, t, labels=None):
        m = self.model
        B = latents.shape[0]
        x_full = m.embed(latents)           # [B, S, d] (replicated latents)
        S = x_full.shape[1]
        assert S % self.world == 0, "num_patches must divide the patch-parallel world"
        P = S // self.world
        x = x_full[:, self.rank * P:(self.rank + 1) * P]
        c = m.cond(t, labels, B, latents.device)
        use_async = self._step >= self.warmup_steps
        for i, blk in enumerate(m.blocks):
            k_loc, v_loc = blk.kv(blk.modulated(x, c))
            if use_async and self._kv_cache[i] is not None:
                # consume last step's gathered K/V, splice in THIS step's
                # local shard (self patches always fresh)
                if self._pending[i] is not None:
                    work, bufs = self._pending[i]
                    work.wait()
                    self._kv_cache[i] = self._assemble(bufs)
                k_full, v_full = self._kv_cache[i]
                k_full = k_full.clone()
                v_full = v_full.clone()
                k_full[:, self.rank * P:(self.rank + 1) * P] = k_loc
                v_full[:, self.rank * P:(self.rank + 1) * P] = v_loc
                self._pending[i] = self._gather_kv(i, k_loc, v_loc, True)
            else:
                work, bufs = self._gather_kv(i, k_loc, v_loc, False)
                k_full, v_full = self._assemble(bufs)
                self._kv_cache[i] = (k_full, v_full)
                self._pending[i] = None
            x = blk(x, c, kv_override=(k_full, v_full))
        shift, scale = m.final_ada(torch.nn.functional.silu(c)).chunk(2, dim=-1)
        x = m.final_norm(x) * (1 + scale[:, None]) + shift[:, None]
        x = m.final_proj(x)
        # gather output patch shards
        outs = [torch.empty_like(x) for _ in range(self.world)]
        dist.all_gather(outs, x.contiguous(), group=self.group)
        full = torch.cat(outs, dim=1)
        self._step += 1
        return {"sample": m.unpatchify(full), "loss": None}


class DiffusionEngine:
    """reference: colossalai/inference/core/diffusion_engine.py — generate()
    returns denoised latents (the VAE/image decode stage is model-external)."""

    def __init__(self, model: DiT, patch_parallel_group=None, warmup_steps: int = 2):
        self.model = model
        if patch_parallel_group is not None and dist.get_world_size(patch_parallel_group) > 1:
            self.model = PatchParallelDiT(model, patch_parallel_group, warmup_steps)
        self.cfg = model.config

    @torch.no_grad()
    def generate(self, num_images: int = 1, labels: Optional[List[int]] = None,
                 steps: int = 50, guidance_scale: float = 4.0, seed: Optional[int] = None):
        base = self.model.model if isinstance(self.model, PatchParallelDiT) else self.model
        device = next(base.parameters()).device
        gen = None
        if seed is not None:
            gen = torch.Generator(device=device)
            gen.manual_seed(seed)
        if isinstance(self.model, PatchParallelDiT):
            self.model.reset()
        lab = torch.tensor(labels, device=device) if labels is not None else None
        shape = (num_images, self.cfg.in_channels, self.cfg.input_size, self.cfg.input_size)
        return ddim_sample(self.model, shape, steps, guidance_scale, lab, generator=gen,
                           device=device)
