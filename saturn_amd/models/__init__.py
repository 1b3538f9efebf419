from .mlp import MLP, get_mlp_dataloader, get_mlp_model, mse_loss

__all__ = ["MLP", "get_mlp_model", "get_mlp_dataloader", "mse_loss"]
