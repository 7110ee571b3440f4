from .manager import BackupError, BackupManager

__all__ = ["BackupError", "BackupManager"]
