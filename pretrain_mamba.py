"""Mamba / hybrid SSM pretraining entry point (reference: pretrain_mamba.py).

  torchrun --nproc-per-node 8 --master-addr 127.0.0.1 pretrain_mamba.py \
      --num-layers 48 --hidden-size 4096 --num-attention-heads 32 \
      --hybrid-attention-ratio 0.08 --hybrid-mlp-ratio 0.5 \
      --seq-length 4096 --micro-batch-size 1 --global-batch-size 64 --bf16 \
      --use-distributed-optimizer --mock-data --train-iters 20
"""

from megatron_amd.models.mamba import MambaModel
from megatron_amd.training.pretrain import pretrain


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    return MambaModel(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)


if __name__ == "__main__":
    pretrain(model_provider)
