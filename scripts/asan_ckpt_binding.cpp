// Minimal binding for the host-ASan/UBSan build of the native checkpoint
// serializer (fmda_amd/ops/csrc/checkpoint.cpp) — sanitizer run only.
#include <torch/extension.h>

#include <string>
#include <utility>
#include <vector>

namespace fmda_ckpt {
void save_state_dict(const std::string& path,
                     const std::vector<std::string>& keys,
                     const std::vector<torch::Tensor>& tensors);
std::vector<std::pair<std::string, torch::Tensor>> load_state_dict(
        const std::string& path);
}  // namespace fmda_ckpt

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("save_state_dict_native", &fmda_ckpt::save_state_dict);
    m.def("load_state_dict_native", &fmda_ckpt::load_state_dict);
}
