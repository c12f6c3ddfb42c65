#!/usr/bin/env python3
"""`bng` entry point: python bng.py run|demo|stats|version"""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from bng_amd.cli.main import main
sys.exit(main())
