"""Multimodal (LLaVA) pretraining entry point (reference: pretrain_vlm.py).

Vision encoder + projector + language model; the image token in each text
sample is spliced with the projected patch embeddings.  Mock mode feeds
synthetic images alongside the synthetic token stream.
"""

import torch

from megatron_amd.models.llava import DEFAULT_IMAGE_TOKEN_INDEX, LLaVAModel
from megatron_amd.training.pretrain import pretrain


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    from dataclasses import replace

    vision_cfg = replace(
        config,
        num_layers=max(2, config.num_layers // 4),
        causal_attention=False,
        position_embedding_type="learned",
    )
    return LLaVAModel(language_config=config, vision_config=vision_cfg,
                      img_h=112, img_w=112, patch_dim=14)


def forward_step_builder(args):
    def forward_step(data_iterator, model):
        batch = next(data_iterator)
        b, s = batch["tokens"].shape
        tokens = batch["tokens"].clone()
        tokens[:, 0] = DEFAULT_IMAGE_TOKEN_INDEX  # one image per sample
        labels = batch["labels"]
        core = model.module if hasattr(model, "module") else model
        g = torch.Generator().manual_seed(args.seed + 1)
        images = torch.randn(b, 3, core.img_h, core.img_w, generator=g).to(
            tokens.device, next(model.parameters()).dtype)

        def loss_func(loss_sb):
            total = loss_sb.sum()
            ntok = torch.tensor(loss_sb.numel(), device=loss_sb.device)
            return total, ntok, {"loss_sum": total.detach()}

        out = model(images=images, input_ids=tokens, labels=labels)
        return out, loss_func

    return forward_step


if __name__ == "__main__":
    pretrain(model_provider, forward_step_builder=forward_step_builder)
