"""The ``ck`` command-line tool (placeholder until the dev loop lands)."""


def main() -> None:  # pragma: no cover
    from .main import run
    run()
