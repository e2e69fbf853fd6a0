"""Build every HIP extension in-tree for gfx950 (driver 'does it build' check)."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

def main():
    env = dict(os.environ)
    env.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')
    env.setdefault('MAX_JOBS', '8')
    subprocess.run([sys.executable, 'setup.py', 'build_ext', '--inplace'],
                   check=True, cwd=REPO, env=env)

if __name__ == '__main__':
    main()
