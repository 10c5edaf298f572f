from .notifications import NotificationManager

__all__ = ["NotificationManager"]
