"""Python interpreter exit-status guard (parity: reference
python/utils/exit_status.py:19-33): lets daemon threads/subprocess loops
distinguish a normal interpreter shutdown from a crash."""
import atexit

_python_exit_status = False


def _mark_exit():
    global _python_exit_status
    _python_exit_status = True


atexit.register(_mark_exit)


def python_exit_status() -> bool:
    return _python_exit_status
