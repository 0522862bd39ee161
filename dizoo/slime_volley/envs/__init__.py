from .slime_volley_env import SlimeVolleyEnv
