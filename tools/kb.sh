#!/bin/bash
cd /root/repo
exec ./bigclam/kernels/kbench "${1:-5000}" "${2:-}"
