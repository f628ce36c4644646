from .main import entry

entry()
