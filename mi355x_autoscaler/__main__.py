"""``python -m mi355x_autoscaler`` == ``python -m mi355x_autoscaler.cli``
(the reference ran ``python main.py`` [R: Dockerfile CMD — reconstructed])."""

from .cli import main

if __name__ == "__main__":
    main()
