"""GPT/Llama pretraining entry point (reference: pretrain_gpt.py).

  torchrun --nproc-per-node 8 --master-addr 127.0.0.1 pretrain_gpt.py \
      --num-layers 32 --hidden-size 4096 --num-attention-heads 32 \
      --num-query-groups 8 --ffn-hidden-size 14336 --seq-length 4096 \
      --micro-batch-size 1 --global-batch-size 64 --bf16 \
      --use-distributed-optimizer --mock-data --train-iters 20
"""

from megatron_amd.models.gpt import GPTModel
from megatron_amd.training.pretrain import pretrain


def model_provider(config, pre_process=True, post_process=True, vp_stage=None):
    return GPTModel(config, pre_process=pre_process, post_process=post_process, vp_stage=vp_stage)


if __name__ == "__main__":
    pretrain(model_provider)
