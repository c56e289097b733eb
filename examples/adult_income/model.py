"""The adult-income DNN (shape mirrors the reference example
examples/src/adult-income/model.py: dense MLP + sparse MLP + 3 FC layers)."""
from typing import List

import torch
import torch.nn as nn


class DNN(nn.Module):
    def __init__(self, dense_mlp_output_size: int = 16, sparse_mlp_output_size: int = 128):
        super().__init__()
        self.dense_mlp = nn.Linear(5, dense_mlp_output_size)
        self.dense_bn = nn.BatchNorm1d(dense_mlp_output_size)
        self.sparse_mlp = nn.Linear(64, sparse_mlp_output_size)  # 8 slots x dim 8
        self.sparse_bn = nn.BatchNorm1d(sparse_mlp_output_size)
        self.ln1 = nn.Linear(dense_mlp_output_size + sparse_mlp_output_size, 256)
        self.ln2 = nn.Linear(256, 128)
        self.ln3 = nn.Linear(128, 1)
        self.sigmoid = nn.Sigmoid()

    def forward(self, non_id_tensors: List[torch.Tensor], embedding_tensors: List[torch.Tensor]):
        dense_x = non_id_tensors[0].float()
        sparse_concat = torch.cat([e.float() for e in embedding_tensors], dim=1)
        sparse = self.sparse_bn(self.sparse_mlp(sparse_concat))
        dense_x = self.dense_bn(self.dense_mlp(dense_x))
        x = torch.cat([sparse, dense_x], dim=1)
        x = self.ln1(x)
        x = self.ln2(x)
        x = self.ln3(x)
        return self.sigmoid(x)
