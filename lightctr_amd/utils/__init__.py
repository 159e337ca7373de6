from .metrics import auc_score, HistAUC, precision_recall_f1

__all__ = ["auc_score", "HistAUC", "precision_recall_f1"]
