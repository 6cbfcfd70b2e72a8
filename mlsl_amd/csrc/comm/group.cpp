#include "group.hpp"

#include <algorithm>

namespace mlsl {

ProcessGroup::ProcessGroup(int uid, std::vector<int> ranks, int my_world_rank)
    : uid_(uid), ranks_(std::move(ranks)) {
    std::sort(ranks_.begin(), ranks_.end());
    auto it = std::find(ranks_.begin(), ranks_.end(), my_world_rank);
    my_idx_ = it == ranks_.end() ? -1 : static_cast<int>(it - ranks_.begin());
}

ProcessGroup::~ProcessGroup() = default;

}  // namespace mlsl
