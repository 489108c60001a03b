"""BERT pretraining entry point (reference: pretrain_bert.py).

Masked-LM over an indexed corpus (or mock data), bidirectional encoder.
"""

from megatron_amd.models.bert import BertModel
from megatron_amd.training.pretrain import pretrain


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    return BertModel(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)


def forward_step_builder(args):
    import torch

    def forward_step(data_iterator, model):
        batch = next(data_iterator)
        if "loss_mask" not in batch or batch["loss_mask"].min() >= 1.0:
            # mock stream: apply dynamic masking on the fly
            from megatron_amd.datasets.bert_dataset import BertMaskedDataset

            tokens = batch["tokens"]
            mask = torch.rand_like(tokens, dtype=torch.float32) < 0.15
            labels = torch.where(mask, tokens, torch.zeros_like(tokens))
            masked = torch.where(mask, torch.full_like(tokens, args.vocab_size - 1), tokens)
            batch = {"tokens": masked, "labels": labels,
                     "loss_mask": mask.float()}

        def loss_func(loss_sb):
            s = loss_sb.sum()
            ntok = batch["loss_mask"].sum().long().clamp(min=1)
            return s, ntok, {"loss_sum": s.detach()}

        out = model(batch["tokens"], labels=batch["labels"], loss_mask=batch["loss_mask"])
        return out, loss_func

    return forward_step


if __name__ == "__main__":
    pretrain(model_provider, forward_step_builder=forward_step_builder)
