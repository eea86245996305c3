// Toy custom op for the ONNX custom-op export tutorial.
// Reference parity: others/deploy/pytorch2onnx/my_add.cpp:5-12 (3a + 2b).
#include <torch/extension.h>

torch::Tensor my_add(torch::Tensor a, torch::Tensor b) {
  return 3 * a + 2 * b;
}
