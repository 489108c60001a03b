"""Audio-language pretraining entry point (reference: audio-in multimodal
models under megatron/core/models/audio).

Audio feature segments (mock: random mel frames) are projected through the
stacked-frame projector and spliced at the audio placeholder token in each
text sample.
"""

import torch

from megatron_amd.models.audio import AudioLanguageModel
from megatron_amd.training.pretrain import pretrain

FEAT_DIM = 80


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    return AudioLanguageModel(config, feat_dim=FEAT_DIM, stack_factor=4)


def forward_step_builder(args):
    def forward_step(data_iterator, model):
        batch = next(data_iterator)
        b, s = batch["tokens"].shape
        tokens = batch["tokens"].clone()
        core = model.module if hasattr(model, "module") else model
        tokens[:, 0] = core.audio_token_index  # one audio segment per sample
        g = torch.Generator().manual_seed(args.seed + 2)
        feats = torch.randn(b, 16, FEAT_DIM, generator=g).to(
            tokens.device, next(model.parameters()).dtype)

        def loss_func(loss_sb):
            total = loss_sb.sum()
            ntok = torch.tensor(loss_sb.numel(), device=loss_sb.device)
            return total, ntok, {"loss_sum": total.detach()}

        out = model(feats, tokens, labels=batch["labels"])
        return out, loss_func

    return forward_step


if __name__ == "__main__":
    pretrain(model_provider, forward_step_builder=forward_step_builder)
