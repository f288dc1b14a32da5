"""``python -m kuberay_amd`` — the operator binary (see kuberay_amd.operator)."""
import sys

from .operator import main

sys.exit(main())
