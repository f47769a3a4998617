"""Sandbox: isolated exec + filesystem surface.

    python examples/sandbox_demo.py   (with B9_GATEWAY/B9_TOKEN set)
"""

import beta9_amd


def main():
    sb = beta9_amd.Sandbox(cpu=1000, memory=1024).create()
    try:
        print(sb.exec("echo hello from $(hostname)")["stdout"])
        print(sb.run_code("print(6 * 7)")["stdout"])
        sb.fs.upload("/tmp/data.txt", b"persisted bytes")
        print(sb.fs.download("/tmp/data.txt"))
    finally:
        sb.terminate()


if __name__ == "__main__":
    main()
