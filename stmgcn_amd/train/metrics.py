"""Evaluation metrics — formula parity with reference Model_Trainer.py:100-114.

Computed on DENORMALIZED predictions (reference Model_Trainer.py:89-90).
MAPE uses the reference's unusually large epsilon=1.0 (quirk 11).
PCC exists in the reference but is never called; kept for capability parity.
"""
from __future__ import annotations

import numpy as np


def MSE(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.mean(np.square(y_pred - y_true)))


def RMSE(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.sqrt(np.mean(np.square(y_pred - y_true))))


def MAE(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.mean(np.abs(y_pred - y_true)))


def MAPE(y_pred: np.ndarray, y_true: np.ndarray, epsilon: float = 1e-0) -> float:
    return float(np.mean(np.abs(y_pred - y_true) / (y_true + epsilon)))


def PCC(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.corrcoef(y_pred.flatten(), y_true.flatten())[0, 1])
