"""Version stamp (reference pkg/version analogue)."""
import subprocess

VERSION = "0.1.0"


def git_commit() -> str:
    try:
        return subprocess.run(["git", "rev-parse", "--short", "HEAD"],
                              capture_output=True, text=True,
                              timeout=5).stdout.strip() or "unknown"
    except Exception:  # noqa: BLE001
        return "unknown"


def user_agent() -> str:
    """pkg/utils/useragent Default analogue."""
    import platform
    return f"lws-amd/{VERSION} ({platform.system().lower()}/{platform.machine()}) {git_commit()}"
