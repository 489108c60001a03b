"""Dense -> MoE upcycling.

Capability analog of reference megatron/core/transformer/moe/
upcycling_utils.py: initialize an MoE model from a trained dense model —
attention / norms / embeddings copied 1:1, each expert's MLP seeded from
the dense MLP (optionally noise-perturbed so experts diverge), router
freshly initialized.  Both models must be built on the same grid (same TP;
EP shards just take their local expert copies, which are identical).
"""

from __future__ import annotations

from typing import Optional

import torch


@torch.no_grad()
def upcycle_dense_to_moe(dense_model: torch.nn.Module, moe_model: torch.nn.Module,
                         noise_std: float = 0.0, scale_fc2: Optional[float] = None) -> int:
    """Copy dense weights into the MoE twin.  `noise_std` adds Gaussian
    perturbation to each expert copy (symmetric experts never diverge under
    identical routing); `scale_fc2` rescales expert outputs (reference uses
    1/topk-style scaling when router probs multiply outputs).  Returns the
    number of upcycled MoE layers."""
    from megatron_amd.moe.moe_layer import MoELayer

    dense_core = dense_model.module if hasattr(dense_model, "module") else dense_model
    moe_core = moe_model.module if hasattr(moe_model, "module") else moe_model

    dense_named = dict(dense_core.named_parameters())

    # 1:1 copy of everything that exists under the same name (embeddings,
    # norms, attention, output layer)
    for name, p in moe_core.named_parameters():
        if name in dense_named and dense_named[name].shape == p.shape:
            p.copy_(dense_named[name])

    # expert seeding: each decoder layer's MoE takes the dense layer's MLP
    n_upcycled = 0
    dense_layers = list(dense_core.decoder.layers)
    moe_layers = list(moe_core.decoder.layers)
    assert len(dense_layers) == len(moe_layers)
    for dl, ml in zip(dense_layers, moe_layers):
        moe = getattr(ml, "mlp", None)
        if not isinstance(moe, MoELayer):
            continue
        dense_mlp = dl.mlp
        fc1 = dense_mlp.linear_fc1.weight  # [fc1_out/tp, h]
        fc2 = dense_mlp.linear_fc2.weight  # [h, ffn/tp]
        experts = moe.experts
        E = experts.weight1.shape[0]
        assert experts.weight1.shape[1:] == fc1.shape, \
            f"expert fc1 {tuple(experts.weight1.shape[1:])} vs dense {tuple(fc1.shape)}: " \
            "upcycling requires moe_ffn_hidden_size == ffn_hidden_size"
        for e in range(E):
            experts.weight1[e].copy_(fc1)
            experts.weight2[e].copy_(fc2 if scale_fc2 is None else fc2 * scale_fc2)
            if noise_std > 0:
                experts.weight1[e].add_(torch.randn_like(fc1) * noise_std)
                experts.weight2[e].add_(torch.randn_like(fc2) * noise_std)
        if moe.shared_expert is not None and \
                moe.shared_expert.linear_fc1.weight.shape == fc1.shape:
            moe.shared_expert.linear_fc1.weight.copy_(fc1)
            moe.shared_expert.linear_fc2.weight.copy_(fc2)
        n_upcycled += 1
    return n_upcycled
