"""T5 pretraining entry point (reference: pretrain_t5.py).

Span corruption over an indexed corpus (or mock data): encoder sees
sentinel-masked text, decoder reconstructs the masked spans
(datasets/t5_dataset.py).
"""

from megatron_amd.models.t5 import T5Model
from megatron_amd.training.pretrain import pretrain


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    return T5Model(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)


def forward_step_builder(args):
    import torch

    from megatron_amd.datasets.t5_dataset import T5SpanCorruptionDataset, pad_t5_batch

    class _RowDataset:
        def __init__(self, tokens):
            self.tokens = tokens

        def __len__(self):
            return self.tokens.shape[0]

        def __getitem__(self, i):
            return {"tokens": self.tokens[i]}

    def forward_step(data_iterator, model):
        raw = next(data_iterator)
        if "encoder_tokens" not in raw:
            # GPT-style token stream -> span-corrupt on the fly
            tokens = raw["tokens"]
            ds = T5SpanCorruptionDataset(
                _RowDataset(tokens), vocab_size=args.vocab_size,
                bos_id=0, eos_id=1, seed=args.seed)
            batch = pad_t5_batch([ds[i] for i in range(len(ds))],
                                 enc_len=args.seq_length,
                                 dec_len=max(args.seq_length // 4, 32))
        else:
            batch = raw

        def loss_func(loss_sb):
            s = loss_sb.sum()
            ntok = batch["loss_mask"].sum().long().clamp(min=1)
            return s, ntok, {"loss_sum": s.detach()}

        out = model(
            encoder_input_ids=batch["encoder_tokens"],
            decoder_input_ids=batch["decoder_tokens"],
            labels=batch["labels"],
            loss_mask=batch["loss_mask"],
        )
        return out, loss_func

    return forward_step


if __name__ == "__main__":
    pretrain(model_provider, forward_step_builder=forward_step_builder)
