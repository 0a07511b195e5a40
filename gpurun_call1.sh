set -x
exec > gpurun_out/call1.log 2>&1
echo "=== device census ==="
ls /dev/kfd /dev/dri/ 2>&1
python -c "import torch; print('torch devices:', torch.cuda.device_count())"
/opt/rocm/bin/amd-smi list 2>&1 | head -40
echo "=== skip-gate diagnostics ==="
ls -la /dev/loop-control 2>&1
cat /sys/fs/cgroup/cgroup.controllers 2>&1
id
echo "can write cgroup?"; mkdir /sys/fs/cgroup/gda-probe 2>&1 && echo yes && rmdir /sys/fs/cgroup/gda-probe
echo "loop mount probe:"
t=$(mktemp -d); truncate -s 16M $t/img; mkfs.ext4 -q -F $t/img && mount -o loop $t/img $t 2>&1 && echo LOOP_OK && umount $t; rm -rf $t
echo "=== gpu pytest ==="
timeout 420 python -m pytest tests -m gpu -q -rs -p no:cacheprovider
echo "pytest rc=$?"
