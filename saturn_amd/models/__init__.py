from .mlp import MLP, get_mlp_dataloader, get_mlp_model, mse_loss
from .gpt2 import GPT2ForCausalLM, get_gpt2_model, gpt2_loss
from .llama import LlamaForCausalLM, get_llama_model, llama_loss
from .mixtral import MixtralForCausalLM, get_mixtral_model, mixtral_loss
from .bert import BertForMaskedLM, get_bert_model, make_mlm_dataloader, mlm_loss
from .vit import ViTForImageClassification, get_vit_model, make_image_dataloader, vit_loss
from .gptj import (
    GPTJForCausalLM,
    get_gptj_model,
    make_token_dataloader,
    pretraining_loss,
)

__all__ = [
    "MLP",
    "get_mlp_model",
    "get_mlp_dataloader",
    "mse_loss",
    "GPT2ForCausalLM",
    "get_gpt2_model",
    "gpt2_loss",
    "GPTJForCausalLM",
    "get_gptj_model",
    "make_token_dataloader",
    "pretraining_loss",
    "LlamaForCausalLM",
    "get_llama_model",
    "MixtralForCausalLM",
    "get_mixtral_model",
    "mixtral_loss",
    "llama_loss",
    "BertForMaskedLM",
    "get_bert_model",
    "make_mlm_dataloader",
    "mlm_loss",
    "ViTForImageClassification",
    "get_vit_model",
    "make_image_dataloader",
    "vit_loss",
]
