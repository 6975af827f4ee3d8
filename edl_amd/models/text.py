"""Text-classification students for NLP distillation (parity: reference
example/distill/nlp/model.py:85-135 — BOW and CNN students distilled from
a served BERT teacher; the teacher here is any served model exposing
logits through the teacher_server).

Behavior-equivalent torch modules:
  BOW:  embedding(128, padding_idx 0) -> pad-masked sum -> softsign -> fc
  CNN:  embedding -> width-3 conv over tokens (relu) -> width-3 avg pool
        -> pad-masked sum -> softsign -> fc
Both return raw logits; the KD loss (ops.functional.kd_soft_cross_entropy)
applies the soft-label CE the reference used (KL_T / soft-label
softmax_with_cross_entropy, nlp/model.py:54-66)."""
import torch
import torch.nn as nn
import torch.nn.functional as F


class TextBOW(nn.Module):
    def __init__(self, vocab_size, num_classes=2, emb_dim=128):
        super().__init__()
        self.emb = nn.Embedding(vocab_size, emb_dim, padding_idx=0)
        self.fc = nn.Linear(emb_dim, num_classes)

    def forward(self, ids):
        e = self.emb(ids)                                  # [B, T, D]
        mask = (ids != 0).unsqueeze(-1).to(e.dtype)        # pad mask
        h = F.softsign((e * mask).sum(1))
        return self.fc(h)


class TextCNN(nn.Module):
    def __init__(self, vocab_size, num_classes=2, emb_dim=128):
        super().__init__()
        self.emb = nn.Embedding(vocab_size, emb_dim, padding_idx=0)
        self.conv = nn.Conv1d(emb_dim, emb_dim, 3, padding=1)
        self.fc = nn.Linear(emb_dim, num_classes)

    def forward(self, ids):
        e = self.emb(ids)                                  # [B, T, D]
        h = F.relu(self.conv(e.transpose(1, 2)))           # [B, D, T]
        h = F.avg_pool1d(h, 3, stride=1, padding=1)
        h = h.transpose(1, 2)                              # [B, T, D]
        mask = (ids != 0).unsqueeze(-1).to(h.dtype)
        h = F.softsign((h * mask).sum(1))
        return self.fc(h)
