"""BERT pretraining entry point (reference: pretrain_bert.py).

Masked-LM over an indexed corpus (or mock data), bidirectional encoder.
"""

from megatron_amd.models.bert import BertModel
from megatron_amd.training.pretrain import pretrain


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    return BertModel(config, pre_process=pre_process, post_process=post_process,
                     vp_stage=vp_stage, add_binary_head=True)


def forward_step_builder(args):
    import torch

    def forward_step(data_iterator, model):
        batch = next(data_iterator)
        if "loss_mask" not in batch or batch["loss_mask"].min() >= 1.0:
            # mock stream: apply dynamic masking on the fly
            from megatron_amd.datasets.bert_dataset import BertMaskedDataset

            tokens = batch["tokens"]
            b, s_len = tokens.shape
            mask = torch.rand_like(tokens, dtype=torch.float32) < 0.15
            labels = torch.where(mask, tokens, torch.zeros_like(tokens))
            masked = torch.where(mask, torch.full_like(tokens, args.vocab_size - 1), tokens)
            # NSP pairs: second segment swapped with a shifted row 50% of the
            # time; tokentype ids mark the segments
            half = s_len // 2
            swap = torch.rand(b) < 0.5
            masked = masked.clone()
            masked[swap, half:] = masked.roll(1, dims=0)[swap, half:]
            tokentype = torch.zeros_like(tokens)
            tokentype[:, half:] = 1
            batch = {"tokens": masked, "labels": labels, "loss_mask": mask.float(),
                     "tokentype_ids": tokentype,
                     "is_next": (~swap).long().to(tokens.device)}

        core = model.module if hasattr(model, "module") else model

        def loss_func(loss_sb):
            s = loss_sb.sum()
            ntok = batch["loss_mask"].sum().long().clamp(min=1)
            if getattr(core, "binary_logits", None) is not None and "is_next" in batch:
                # NSP loss joins MLM, scaled per token so the schedule's
                # 1/ntok normalization applies once (reference pretrain_bert)
                nsp = torch.nn.functional.cross_entropy(
                    core.binary_logits.float(), batch["is_next"])
                s = s + nsp * ntok
            return s, ntok, {"loss_sum": s.detach()}

        out = model(batch["tokens"], tokentype_ids=batch.get("tokentype_ids"),
                    labels=batch["labels"], loss_mask=batch["loss_mask"])
        return out, loss_func

    return forward_step


if __name__ == "__main__":
    pretrain(model_provider, forward_step_builder=forward_step_builder)
