#!/bin/bash
# amdgpu driver + ROCm userspace installer for minikube nodes.
#
# Role parity: /root/reference/nvidia-driver-installer/minikube/entrypoint.sh —
# same flow as the Ubuntu installer plus a kernel-source fallback
# (minikube/entrypoint.sh:35-56 downloads from cdn.kernel.org with version
# fixups) because minikube VM kernels ship without a matching
# linux-headers package.  MI355X redesign: amdgpu-dkms builds against the
# prepared source tree; no proprietary .run installer, no overlayfs trick.
set -o errexit
set -o pipefail
set -u

set -x

ROCM_VERSION="${ROCM_VERSION:-7.2}"
AMDGPU_DRIVER_VERSION="${AMDGPU_DRIVER_VERSION:-30.20}"
ROOT_MOUNT_DIR="${ROOT_MOUNT_DIR:-/root}"
INSTALL_DIR_HOST="${INSTALL_DIR_HOST:-/home/kubernetes/bin/amd}"
INSTALL_DIR="${ROOT_MOUNT_DIR}${INSTALL_DIR_HOST}"
CACHE_FILE="${INSTALL_DIR}/.cache"
KERNEL_VERSION="$(uname -r)"
KERNEL_SRC_DIR="/usr/src/linux-${KERNEL_VERSION}"

check_cached_version() {
  [[ -f "${CACHE_FILE}" ]] || return 1
  grep -q "^CACHE_KERNEL_VERSION=${KERNEL_VERSION}$" "${CACHE_FILE}" || return 1
  grep -q "^CACHE_AMDGPU_VERSION=${AMDGPU_DRIVER_VERSION}$" "${CACHE_FILE}" || return 1
  grep -q "^CACHE_ROCM_VERSION=${ROCM_VERSION}$" "${CACHE_FILE}" || return 1
  echo "amdgpu ${AMDGPU_DRIVER_VERSION} + ROCm ${ROCM_VERSION} already installed for ${KERNEL_VERSION}"
  return 0
}

update_cached_version() {
  cat >"${CACHE_FILE}" <<EOF
CACHE_KERNEL_VERSION=${KERNEL_VERSION}
CACHE_AMDGPU_VERSION=${AMDGPU_DRIVER_VERSION}
CACHE_ROCM_VERSION=${ROCM_VERSION}
EOF
}

prepare_kernel_source() {
  # Minikube-specific: no linux-headers package for the VM kernel, so fetch
  # the matching stable source and prepare it for out-of-tree (dkms) builds.
  # Parity: minikube/entrypoint.sh:35-56 (cdn.kernel.org + version fixups).
  if apt-get install -y "linux-headers-${KERNEL_VERSION}" 2>/dev/null; then
    return 0
  fi
  local base="${KERNEL_VERSION%%-*}"        # e.g. 5.10.57-generic -> 5.10.57
  local major="${base%%.*}"                 # -> 5
  # kernel.org drops a trailing ".0" from x.y.0 tarball names.
  local tarball="linux-${base%.0}.tar.xz"
  apt-get update
  apt-get install -y wget xz-utils bc bison flex libelf-dev libssl-dev
  wget -q "https://cdn.kernel.org/pub/linux/kernel/v${major}.x/${tarball}" \
    -O /tmp/linux-src.tar.xz
  mkdir -p "${KERNEL_SRC_DIR}"
  tar -xf /tmp/linux-src.tar.xz -C "${KERNEL_SRC_DIR}" --strip-components=1
  pushd "${KERNEL_SRC_DIR}"
  # Configure exactly like the running kernel so dkms modules load.
  if [[ -f "/proc/config.gz" ]]; then
    zcat /proc/config.gz > .config
  elif [[ -f "${ROOT_MOUNT_DIR}/boot/config-${KERNEL_VERSION}" ]]; then
    cp "${ROOT_MOUNT_DIR}/boot/config-${KERNEL_VERSION}" .config
  else
    make defconfig
  fi
  # The source tree must report the VM's exact release string.
  sed -i "s/^EXTRAVERSION.*/EXTRAVERSION = -${KERNEL_VERSION#*-}/" Makefile
  make olddefconfig
  make modules_prepare
  popd
  ln -sfn "${KERNEL_SRC_DIR}" "/lib/modules/${KERNEL_VERSION}/build"
}

setup_repos() {
  local ub_codename
  ub_codename="$(. /etc/os-release && echo "${VERSION_CODENAME}")"
  mkdir -p /etc/apt/keyrings
  curl -fsSL https://repo.radeon.com/rocm/rocm.gpg.key | \
    gpg --dearmor -o /etc/apt/keyrings/rocm.gpg
  echo "deb [arch=amd64 signed-by=/etc/apt/keyrings/rocm.gpg] \
https://repo.radeon.com/amdgpu/${AMDGPU_DRIVER_VERSION}/ubuntu ${ub_codename} main" \
    > /etc/apt/sources.list.d/amdgpu.list
  echo "deb [arch=amd64 signed-by=/etc/apt/keyrings/rocm.gpg] \
https://repo.radeon.com/rocm/apt/${ROCM_VERSION} ${ub_codename} main" \
    > /etc/apt/sources.list.d/rocm.list
  apt-get update
}

build_and_load_kmd() {
  DEBIAN_FRONTEND=noninteractive apt-get install -y amdgpu-dkms
  dkms autoinstall -k "${KERNEL_VERSION}"
  modprobe amdgpu
}

install_rocm_userspace() {
  DEBIAN_FRONTEND=noninteractive apt-get install -y \
    rocm-core rocminfo rocm-smi-lib amd-smi-lib \
    hip-runtime-amd rocblas hipblaslt miopen-hip rccl
  mkdir -p "${INSTALL_DIR}/lib64" "${INSTALL_DIR}/bin" "${INSTALL_DIR}/.info"
  cp -a /opt/rocm/lib/*.so* "${INSTALL_DIR}/lib64/" 2>/dev/null || true
  for tool in rocminfo rocm-smi amd-smi; do
    cp -a "/opt/rocm/bin/${tool}" "${INSTALL_DIR}/bin/" 2>/dev/null || true
  done
  echo "${ROCM_VERSION}" > "${INSTALL_DIR}/.info/rocm-version"
}

verify_installation() {
  # Parity: ubuntu/entrypoint.sh verify step (nvidia-smi analog).
  "${INSTALL_DIR}/bin/rocminfo" | grep "gfx950" > /dev/null
  [[ -e /dev/kfd ]]
  compgen -G "/dev/dri/renderD*" > /dev/null
}

update_host_ld_cache() {
  echo "${INSTALL_DIR_HOST}/lib64" \
    > "${ROOT_MOUNT_DIR}/etc/ld.so.conf.d/amd.conf"
  ldconfig -r "${ROOT_MOUNT_DIR}"
}

main() {
  if check_cached_version; then
    exit 0
  fi
  setup_repos
  prepare_kernel_source
  build_and_load_kmd
  install_rocm_userspace
  verify_installation
  update_host_ld_cache
  update_cached_version
  echo "amdgpu + ROCm install complete for minikube node"
}

# Source-guard: tests may source this file and drive functions directly.
if [[ "${BASH_SOURCE[0]:-}" == "$0" ]]; then
  main "$@"
fi
