from .history import ContentBase, History
